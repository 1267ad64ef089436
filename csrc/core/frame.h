// frame.h — orthonormal frames, local<->world transforms, quaternions.
//
// Capability parity: reference src/core/so3.cuh (rotation_fixed_anchor,
// delocalize_rotate) + quaternion.cuh.  We use the branchless
// Duff/Burgess/Christensen et al. "building an orthonormal basis" construction
// instead of the reference's SO3 matrices — cheaper in registers on wave64.
#pragma once
#include "vec.h"

namespace hippt {

// Orthonormal basis around unit normal n. t/b/n right-handed.
struct Frame {
    Vec3 t, b, n;

    HD static Frame from_n(const Vec3& n) {
        Frame f;
        f.n = n;
        float sign = copysignf(1.f, n.z);
        const float a = -1.f / (sign + n.z);
        const float bb = n.x * n.y * a;
        f.t = Vec3(1.f + sign * n.x * n.x * a, sign * bb, -sign * n.x);
        f.b = Vec3(bb, sign + n.y * n.y * a, -n.y);
        return f;
    }
    // local (z-up) -> world
    HD Vec3 to_world(const Vec3& v) const { return t * v.x + b * v.y + n * v.z; }
    // world -> local
    HD Vec3 to_local(const Vec3& v) const { return {t.dot(v), b.dot(v), n.dot(v)}; }
};

// delocalize_rotate parity: take a local-hemisphere sample (z-up) to the
// hemisphere around `anchor`.
HD Vec3 delocalize(const Vec3& anchor, const Vec3& local) {
    return Frame::from_n(anchor).to_world(local);
}

struct Quat {
    float w, x, y, z;
    HD Quat() : w(1), x(0), y(0), z(0) {}
    HD Quat(float w_, float x_, float y_, float z_) : w(w_), x(x_), y(y_), z(z_) {}
    HD static Quat angle_axis(float rad, const Vec3& axis) {
        float h = 0.5f * rad, s = sinf(h);
        Vec3 a = axis.normalized();
        return {cosf(h), a.x * s, a.y * s, a.z * s};
    }
    HD Quat operator*(const Quat& o) const {
        return {w * o.w - x * o.x - y * o.y - z * o.z,
                w * o.x + x * o.w + y * o.z - z * o.y,
                w * o.y - x * o.z + y * o.w + z * o.x,
                w * o.z + x * o.y - y * o.x + z * o.w};
    }
    HD Vec3 rotate(const Vec3& v) const {
        Vec3 u{x, y, z};
        Vec3 uv = u.cross(v);
        return v + (uv * w + u.cross(uv)) * 2.f;
    }
    HD Quat conj() const { return {w, -x, -y, -z}; }
};

// 3x3 rotation matrix (row-major) — camera pose (reference camera SO3).
struct Mat3 {
    Vec3 r0, r1, r2;  // rows
    HD Mat3() : r0(1, 0, 0), r1(0, 1, 0), r2(0, 0, 1) {}
    HD Mat3(const Vec3& a, const Vec3& b, const Vec3& c) : r0(a), r1(b), r2(c) {}
    HD Vec3 operator*(const Vec3& v) const { return {r0.dot(v), r1.dot(v), r2.dot(v)}; }
    // transpose-multiply (inverse for rotation)
    HD Vec3 t_mul(const Vec3& v) const {
        return {fmaf(r0.x, v.x, fmaf(r1.x, v.y, r2.x * v.z)),
                fmaf(r0.y, v.x, fmaf(r1.y, v.y, r2.y * v.z)),
                fmaf(r0.z, v.x, fmaf(r1.z, v.y, r2.z * v.z))};
    }
    // columns = camera basis: right, up, forward
    HD static Mat3 from_cols(const Vec3& c0, const Vec3& c1, const Vec3& c2) {
        return Mat3({c0.x, c1.x, c2.x}, {c0.y, c1.y, c2.y}, {c0.z, c1.z, c2.z});
    }
};

} // namespace hippt
