// bvh.h — linearized BVH node layout + stackless device/host traversal.
//
// Capability parity: reference src/core/bvh.cuh (LinearNode, 2 float4/node,
// negative-offset skip jumps) + src/renderer/tracing_func.cuh:44-181
// (ray_intersect_bvh / occlusion_test_bvh).
//
// MI355X-native design: nodes are stored in DFS order with an explicit
// skip-link (the reference encodes the same thing as a negative subtree
// offset).  Traversal is a single while loop with no stack, no texture
// fetches (CDNA has no texture-cache path worth using for this): two 16-byte
// vector loads per node, branchless index advance, early-out on tmax.  The
// megakernel caches the top of the tree in LDS (see pt_kernels.hip).
#pragma once
#include "geometry.h"

namespace hippt {

// Node = 32 bytes:
//   lo.xyz / hi.xyz  — AABB
//   lo.w  (int)      — leaf: first primitive index;  internal: unused (-1)
//   hi.w  (int)      — leaf: primitive count (>0);   internal: -skip_index
// "skip_index" = node index of the next subtree in DFS order (where to jump
// on an AABB miss).  For a leaf the skip target is node_idx+1, so only
// internal nodes store it.
struct alignas(16) BVHNode {
    Vec4 lo;
    Vec4 hi;

    HD bool is_leaf() const { return float_as_int(hi.w) > 0; }
    HD int prim_base() const { return float_as_int(lo.w); }
    HD int prim_cnt() const { return float_as_int(hi.w); }
    HD int skip() const { return -float_as_int(hi.w); }
    HD AABB aabb() const { return AABB(lo.xyz(), hi.xyz()); }
};

// Safe direction reciprocal: a zero (or denormal) direction component makes
// inv_d infinite and the slab test compute 0*inf = NaN; under -ffast-math
// fmaxf(NaN, 0) = 0, so every AABB tests as HIT and one ray scans the whole
// tree (measured: 90 ms dispatches vs 1 ms median on MI355X).  Clamping the
// component magnitude keeps inv_d finite and the test conservative.
HD float safe_rcp_1(float v) {
    const float eps = 1e-12f;
    // bit-level NaN/Inf test (survives -ffast-math where v != v folds away)
    uint32_t bits = float_as_uint(v);
    if ((bits & 0x7f800000u) == 0x7f800000u) v = eps;
    if (fabsf(v) < eps) v = copysignf(eps, v);
    return 1.f / v;
}
HD Vec3 safe_rcp_dir(const Vec3& d) {
    return {safe_rcp_1(d.x), safe_rcp_1(d.y), safe_rcp_1(d.z)};
}

struct HitRecord {
    float t;
    float u, v;
    int prim_idx;   // -1 = miss
    HD HitRecord() : t(MAX_DIST), u(0), v(0), prim_idx(-1) {}
};

// Closest-hit traversal.  `prim_obj[i]` carries PRIM_SPHERE_BIT.
//
// A speculative dual-fetch variant (load both successors per step through a
// sentinel row) was measured 18% SLOWER on MI355X across megakernel and
// wavefront (extra fetch traffic + register pressure beat the added MLP;
// profiles/README.md), so the walk stays plain: one 32-byte node load per
// step, branchless index advance.
HD HitRecord ray_intersect_bvh(const BVHNode* nodes, int n_nodes,
                               const Prim* prims, const uint32_t* prim_obj,
                               const Ray& ray, float tmax = MAX_DIST) {
    HitRecord rec;
    rec.t = tmax;
    Vec3 inv_d = safe_rcp_dir(ray.d);
    Vec3 o_div = ray.o * inv_d;
    int i = 0;
    while (i < n_nodes) {
        const BVHNode nd = nodes[i];
        int cnt = float_as_int(nd.hi.w);
        float t_near;
        bool hit_box = nd.aabb().intersect(inv_d, o_div, rec.t, t_near);
        if (hit_box & (cnt > 0)) {
            int base = float_as_int(nd.lo.w);
            for (int k = 0; k < cnt; ++k) {
                int pid = base + k;
                bool sph = (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
                float u, v;
                float t = intersect_prim(prims[pid], sph, ray, u, v);
                if (t > EPSILON && t < rec.t) {
                    rec.t = t; rec.u = u; rec.v = v; rec.prim_idx = pid;
                }
            }
        }
        i = (hit_box | (cnt > 0)) ? i + 1 : -cnt;
    }
    if (rec.prim_idx < 0) rec.t = MAX_DIST;
    return rec;
}

// Any-hit occlusion test: returns true if something blocks [EPSILON, tmax].
HD bool occlusion_test_bvh(const BVHNode* nodes, int n_nodes,
                           const Prim* prims, const uint32_t* prim_obj,
                           const Ray& ray, float tmax) {
    Vec3 inv_d = safe_rcp_dir(ray.d);
    Vec3 o_div = ray.o * inv_d;
    int i = 0;
    while (i < n_nodes) {
        const BVHNode nd = nodes[i];
        int cnt = float_as_int(nd.hi.w);
        float t_near;
        bool hit_box = nd.aabb().intersect(inv_d, o_div, tmax, t_near);
        if (hit_box & (cnt > 0)) {
            int base = float_as_int(nd.lo.w);
            for (int k = 0; k < cnt; ++k) {
                int pid = base + k;
                bool sph = (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
                float u, v;
                float t = intersect_prim(prims[pid], sph, ray, u, v);
                if (t > EPSILON && t < tmax) return true;
            }
        }
        i = (hit_box | (cnt > 0)) ? i + 1 : -cnt;
    }
    return false;
}

} // namespace hippt
