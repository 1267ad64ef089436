// geometry.h — Ray, AABB, primitive records, intersection & interaction.
//
// Capability parity: reference src/core/ray.cuh, aabb.cuh, primitives.cuh,
// aos.cuh (PrecomputedArray).  Layout re-designed for MI355X: each primitive
// is 3 16-byte vectors {v0|flag, e1, e2} so a leaf test is three
// global_load_dwordx4; spheres pack {center,radius} into v0 with a tag bit in
// the object-index array (bit31), matching the reference's packing semantics
// (scene.cu:902-930) without its texture-memory path.
#pragma once
#include "vec.h"

namespace hippt {

struct Ray {
    Vec3 o;
    Vec3 d;
    HD Ray() {}
    HD Ray(const Vec3& o_, const Vec3& d_) : o(o_), d(d_) {}
    HD Vec3 at(float t) const { return fmadd(d, t, o); }
};

struct alignas(16) AABB {
    Vec3 lo; float pad0;
    Vec3 hi; float pad1;
    HD AABB() : lo(1e30f), pad0(0), hi(-1e30f), pad1(0) {}
    HD AABB(const Vec3& l, const Vec3& h) : lo(l), pad0(0), hi(h), pad1(0) {}
    HD void grow(const Vec3& p) { lo = lo.minv(p); hi = hi.maxv(p); }
    HD void grow(const AABB& o) { lo = lo.minv(o.lo); hi = hi.maxv(o.hi); }
    HD Vec3 extent() const { return hi - lo; }
    HD Vec3 centroid() const { return (lo + hi) * 0.5f; }
    HD float area() const {
        Vec3 e = extent();
        if (e.x < 0.f) return 0.f;
        return 2.f * (e.x * e.y + e.y * e.z + e.z * e.x);
    }
    HD bool valid() const { return lo.x <= hi.x; }
    // overlap area between two boxes (SBVH overlap penalty, aabb.cuh:intersection_area)
    HD static float intersection_area(const AABB& a, const AABB& b) {
        Vec3 l = a.lo.maxv(b.lo), h = a.hi.minv(b.hi);
        Vec3 e = h - l;
        if (e.x <= 0.f || e.y <= 0.f || e.z <= 0.f) return 0.f;
        return 2.f * (e.x * e.y + e.y * e.z + e.z * e.x);
    }
    // slab test with precomputed inv dir; returns entry distance (>= 0 clamp) or miss
    HD bool intersect(const Vec3& inv_d, const Vec3& o_div, float tmax, float& t_near) const {
        // o_div = o * inv_d precomputed (reference aabb.cuh:58-77)
        Vec3 t0 = fmadd(lo, inv_d, -o_div);
        Vec3 t1 = fmadd(hi, inv_d, -o_div);
        Vec3 tmin = t0.minv(t1), tmaxv = t0.maxv(t1);
        float enter = fmaxf(tmin.max_elem(), 0.f);
        float exit_ = fminf(tmaxv.min_elem(), tmax);
        t_near = enter;
        return enter <= exit_;
    }
};
HD Vec3 fmadd(const Vec3& a, const Vec3& b, const Vec3& c);

// One primitive = 48 bytes. Triangle: v0 + edges e1=v1-v0, e2=v2-v0.
// Sphere: v0 = {cx,cy,cz,r}; e1/e2 unused.
struct alignas(16) Prim {
    Vec4 v0;  // .w: unused (padding / future flags)
    Vec4 e1;
    Vec4 e2;
};

// Per-primitive shading attributes (normals + uv), 3 vertices.
struct alignas(16) PrimAttr {
    Vec4 n0, n1, n2;        // .w of n0/n1/n2: uv u0,v0,u1
    Vec4 uvrest;            // x=v1, y=u2, z=v2, w unused
};

struct Interaction {
    Vec3 shading_n;
    Vec2 uv;
};

// --- intersection ----------------------------------------------------------

// Moller-Trumbore triangle test. Returns hit distance or -1; fills bary u,v.
HD float intersect_triangle(const Prim& p, const Ray& r, float& u, float& v) {
    Vec3 e1 = p.e1.xyz(), e2 = p.e2.xyz();
    Vec3 pv = r.d.cross(e2);
    float det = e1.dot(pv);
    // two-sided test
    if (fabsf(det) < 1e-12f) return -1.f;
    float inv_det = 1.f / det;
    Vec3 tv = r.o - p.v0.xyz();
    u = tv.dot(pv) * inv_det;
    Vec3 qv = tv.cross(e1);
    v = r.d.dot(qv) * inv_det;
    if (u < 0.f || v < 0.f || u + v > 1.f) return -1.f;
    float t = e2.dot(qv) * inv_det;
    return t;
}

HD float intersect_sphere(const Prim& p, const Ray& r) {
    Vec3 c = p.v0.xyz();
    float rad = p.v0.w;
    Vec3 oc = r.o - c;
    float b = oc.dot(r.d);
    float cc = oc.length2() - rad * rad;
    float disc = b * b - cc;
    if (disc < 0.f) return -1.f;
    float s = sqrtf(disc);
    float t0 = -b - s, t1 = -b + s;
    return t0 > EPSILON ? t0 : t1;
}

// sphere flag lives in bit 31 of the prim->object index array
constexpr uint32_t PRIM_SPHERE_BIT = 0x80000000u;
constexpr uint32_t PRIM_OBJ_MASK   = 0x000FFFFFu;

HD float intersect_prim(const Prim& p, bool is_sphere, const Ray& r, float& u, float& v) {
    if (is_sphere) { u = 0.f; v = 0.f; return intersect_sphere(p, r); }
    return intersect_triangle(p, r, u, v);
}

// Barycentric-interpolated shading normal + uv (reference primitives.cuh get_interaction)
HD Interaction get_interaction(const Prim& p, const PrimAttr& a, bool is_sphere,
                               const Vec3& hit_pos, float u, float v) {
    Interaction it;
    if (is_sphere) {
        Vec3 n = (hit_pos - p.v0.xyz()).normalized();
        it.shading_n = n;
        it.uv = Vec2(0.5f + atan2f(n.y, n.x) * (0.5f / PI), 0.5f + asinf(clampv(n.z, -1.f, 1.f)) * (1.f / PI));
    } else {
        float w = 1.f - u - v;
        Vec3 n = a.n0.xyz() * w + a.n1.xyz() * u + a.n2.xyz() * v;
        float l2 = n.length2();
        if (l2 > 1e-20f) n = n * (1.f / sqrtf(l2));
        else n = p.e1.xyz().cross(p.e2.xyz()).normalized();
        it.shading_n = n;
        Vec2 uv0{a.n0.w, a.n1.w}, uv1{a.n2.w, a.uvrest.x}, uv2{a.uvrest.y, a.uvrest.z};
        it.uv = uv0 * w + uv1 * u + uv2 * v;
    }
    return it;
}

// Geometric normal of a primitive at hit point.
HD Vec3 geometric_normal(const Prim& p, bool is_sphere, const Vec3& hit_pos) {
    if (is_sphere) return (hit_pos - p.v0.xyz()).normalized();
    return p.e1.xyz().cross(p.e2.xyz()).normalized();
}

HD float prim_area(const Prim& p, bool is_sphere) {
    if (is_sphere) return 4.f * PI * p.v0.w * p.v0.w;
    return 0.5f * p.e1.xyz().cross(p.e2.xyz()).length();
}

} // namespace hippt
