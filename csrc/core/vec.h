// vec.h — Vec2/Vec3/Vec4 math for the hippt core (host+device single source).
//
// Capability parity: reference src/core/vec2.cuh / vec3.cuh / vec4.cuh.
// Unlike the reference (float4-backed with FLOAT4() reinterpret macros tied to
// CUDA vector types), this is a plain 16-byte-aligned struct usable from both
// g++-compiled host code and hipcc device code; hot kernels load it with
// 16-byte vector loads (the alignment guarantees a single global_load_dwordx4
// on gfx950).
#pragma once
#include "hd.h"

namespace hippt {

struct Vec2 {
    float x, y;
    HD Vec2() : x(0), y(0) {}
    HD Vec2(float a, float b) : x(a), y(b) {}
    HD explicit Vec2(float a) : x(a), y(a) {}
    HD Vec2 operator+(Vec2 o) const { return {x + o.x, y + o.y}; }
    HD Vec2 operator-(Vec2 o) const { return {x - o.x, y - o.y}; }
    HD Vec2 operator*(float s) const { return {x * s, y * s}; }
    HD Vec2 operator*(Vec2 o) const { return {x * o.x, y * o.y}; }
};

struct alignas(16) Vec4;

struct Vec3 {
    float x, y, z;
    HD Vec3() : x(0), y(0), z(0) {}
    HD Vec3(float a, float b, float c) : x(a), y(b), z(c) {}
    HD explicit Vec3(float a) : x(a), y(a), z(a) {}

    HD Vec3 operator+(const Vec3& o) const { return {x + o.x, y + o.y, z + o.z}; }
    HD Vec3 operator-(const Vec3& o) const { return {x - o.x, y - o.y, z - o.z}; }
    HD Vec3 operator-() const { return {-x, -y, -z}; }
    HD Vec3 operator*(const Vec3& o) const { return {x * o.x, y * o.y, z * o.z}; }
    HD Vec3 operator*(float s) const { return {x * s, y * s, z * s}; }
    HD Vec3 operator/(const Vec3& o) const { return {x / o.x, y / o.y, z / o.z}; }
    HD Vec3 operator/(float s) const { float r = 1.f / s; return {x * r, y * r, z * r}; }
    HD Vec3& operator+=(const Vec3& o) { x += o.x; y += o.y; z += o.z; return *this; }
    HD Vec3& operator-=(const Vec3& o) { x -= o.x; y -= o.y; z -= o.z; return *this; }
    HD Vec3& operator*=(const Vec3& o) { x *= o.x; y *= o.y; z *= o.z; return *this; }
    HD Vec3& operator*=(float s) { x *= s; y *= s; z *= s; return *this; }

    HD float operator[](int i) const { return i == 0 ? x : (i == 1 ? y : z); }
    HD void set(int i, float v) { if (i == 0) x = v; else if (i == 1) y = v; else z = v; }

    HD float dot(const Vec3& o) const { return fmaf(x, o.x, fmaf(y, o.y, z * o.z)); }
    HD Vec3 cross(const Vec3& o) const {
        return {fmaf(y, o.z, -z * o.y), fmaf(z, o.x, -x * o.z), fmaf(x, o.y, -y * o.x)};
    }
    HD float length2() const { return dot(*this); }
    HD float length() const { return sqrtf(length2()); }
    HD Vec3 normalized() const {
#if HIPPT_ON_DEVICE
        float inv = __frsqrt_rn(length2());
#else
        float inv = 1.f / sqrtf(length2());
#endif
        return *this * inv;
    }
    HD Vec3 abs_() const { return {fabsf(x), fabsf(y), fabsf(z)}; }
    HD Vec3 rcp() const { return {1.f / x, 1.f / y, 1.f / z}; }
    HD float max_elem() const { return fmaxf(x, fmaxf(y, z)); }
    HD float min_elem() const { return fminf(x, fminf(y, z)); }
    HD float mean() const { return (x + y + z) * (1.f / 3.f); }
    HD bool is_zero() const { return x == 0.f && y == 0.f && z == 0.f; }
    HD Vec3 maxv(const Vec3& o) const { return {fmaxf(x, o.x), fmaxf(y, o.y), fmaxf(z, o.z)}; }
    HD Vec3 minv(const Vec3& o) const { return {fminf(x, o.x), fminf(y, o.y), fminf(z, o.z)}; }
    HD Vec3 expv() const { return {expf(x), expf(y), expf(z)}; }
    // scrub NaN/Inf (reference Vec4::numeric_err scrubbing); bit test so it
    // survives -ffast-math and resolves identically on host and device
    HD bool has_nan() const {
        uint32_t a, bb, c;
        std::memcpy(&a, &x, 4); std::memcpy(&bb, &y, 4); std::memcpy(&c, &z, 4);
        return ((a & 0x7f800000u) == 0x7f800000u) || ((bb & 0x7f800000u) == 0x7f800000u) ||
               ((c & 0x7f800000u) == 0x7f800000u);
    }
};

HD Vec3 operator*(float s, const Vec3& v) { return v * s; }
HD Vec3 fmadd(const Vec3& a, float b, const Vec3& c) {
    return {fmaf(a.x, b, c.x), fmaf(a.y, b, c.y), fmaf(a.z, b, c.z)};
}
HD Vec3 fmadd(const Vec3& a, const Vec3& b, const Vec3& c) {
    return {fmaf(a.x, b.x, c.x), fmaf(a.y, b.y, c.y), fmaf(a.z, b.z, c.z)};
}
HD Vec3 lerp(const Vec3& a, const Vec3& b, float t) { return a + (b - a) * t; }

struct alignas(16) Vec4 {
    float x, y, z, w;
    HD Vec4() : x(0), y(0), z(0), w(0) {}
    HD Vec4(float a, float b, float c, float d) : x(a), y(b), z(c), w(d) {}
    HD Vec4(const Vec3& v, float d) : x(v.x), y(v.y), z(v.z), w(d) {}
    HD explicit Vec4(float a) : x(a), y(a), z(a), w(a) {}
    HD Vec3 xyz() const { return {x, y, z}; }
    HD Vec4 operator+(const Vec4& o) const { return {x + o.x, y + o.y, z + o.z, w + o.w}; }
    HD Vec4 operator-(const Vec4& o) const { return {x - o.x, y - o.y, z - o.z, w - o.w}; }
    HD Vec4 operator*(float s) const { return {x * s, y * s, z * s, w * s}; }
    HD Vec4& operator+=(const Vec4& o) { x += o.x; y += o.y; z += o.z; w += o.w; return *this; }
};

} // namespace hippt
