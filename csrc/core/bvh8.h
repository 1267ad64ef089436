// bvh8.h — 8-wide BVH node + while-while LDS-stack traversal.
//
// Same design as bvh4.h taken one step further: a 256-byte node (4 cache
// lines) holds 8 child AABBs in SoA layout, so one step of the dependent
// walk issues 16 independent 16-byte loads and tests 8 boxes.  At the
// measured BVH4 operating point (VALUBusy 42%, VALUUtilization 20%) the
// extra slab VALU work is free; what the wider node buys is another
// halving of dependent steps per ray and more uniform per-lane iteration
// counts (wave64 divergence).  Traversal is the same while-while
// phase-batched form with (t_near | tagged-entry) stack entries.
#pragma once
#include "bvh4.h"

namespace hippt {

struct alignas(16) BVH8Node {
    float lo_x[8], lo_y[8], lo_z[8];
    float hi_x[8], hi_y[8], hi_z[8];
    int32_t child[8];   // >=0 internal; <0 leaf (prim_base = ~child); cnt==0 -> empty
    int32_t cnt[8];
};
static_assert(sizeof(BVH8Node) == 256, "BVH8Node must be 256 bytes");

// Closest-hit while-while walk over the 8-wide tree.  Entry encoding as in
// bvh4.h ww: low word bit31 = leaf, [30:27] = prim count, [26:0] = base.
HD HitRecord ray_intersect_bvh8_ww(const BVH8Node* nodes,
                                   const Prim* prims, const uint32_t* prim_obj,
                                   const Ray& ray, float tmax,
                                   uint64_t* lds_slot = nullptr, int lds_n = 0) {
    HitRecord rec;
    rec.t = tmax;
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    uint64_t stack[BVH4_STACK];
    int sp = 0;
    constexpr uint32_t DONE = 0x7fffffffu;
    uint32_t cur = 0;
    for (;;) {
        while (cur < 0x80000000u && cur != DONE) {
            const BVH8Node nd = nodes[cur];
            uint32_t keys[8];
            int nhit = 0;
#pragma unroll
            for (int c = 0; c < 8; ++c) {
                float t0x = fmaf(nd.lo_x[c], inv_d.x, -o_div.x);
                float t1x = fmaf(nd.hi_x[c], inv_d.x, -o_div.x);
                float t0y = fmaf(nd.lo_y[c], inv_d.y, -o_div.y);
                float t1y = fmaf(nd.hi_y[c], inv_d.y, -o_div.y);
                float t0z = fmaf(nd.lo_z[c], inv_d.z, -o_div.z);
                float t1z = fmaf(nd.hi_z[c], inv_d.z, -o_div.z);
                float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                    fmaxf(fminf(t0z, t1z), 0.f));
                float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                    fminf(fmaxf(t0z, t1z), rec.t));
                if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~7u) | (uint32_t)c;
            }
            // insertion sort (nhit <= 8), nearest first
            for (int i = 1; i < nhit; ++i) {
                uint32_t k = keys[i];
                int j = i - 1;
                while (j >= 0 && keys[j] > k) { keys[j + 1] = keys[j]; --j; }
                keys[j + 1] = k;
            }
            uint32_t next = DONE;
            for (int k = nhit - 1; k >= 0; --k) {  // far -> near so near pops first
                int c = (int)(keys[k] & 7u);
                int ch = nd.child[c];
                int pc = nd.cnt[c];
                if (ch < 0 && pc == 0) continue;   // empty slot
                uint32_t lo = ch < 0
                    ? (0x80000000u | ((uint32_t)pc << 27) | (uint32_t)(~ch))
                    : (uint32_t)ch;
                if (k == 0) {
                    next = lo;
                } else {
                    uint64_t e = ((uint64_t)(keys[k] & ~7u) << 32) | lo;
                    if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = e;
                    else stack[sp - lds_n] = e;
                    ++sp;
                }
            }
            if (next != DONE) { cur = next; continue; }
            for (;;) {
                if (sp == 0) { cur = DONE; break; }
                --sp;
                uint64_t e = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
                if (uint_as_float((uint32_t)(e >> 32)) < rec.t) { cur = (uint32_t)e; break; }
            }
        }
        if (cur == DONE) break;
        while (cur >= 0x80000000u) {
            bvh4_leaf_hit(prims, prim_obj, ray, (int)(cur & 0x07ffffffu),
                          (int)((cur >> 27) & 0xfu), rec);
            for (;;) {
                if (sp == 0) { cur = DONE; break; }
                --sp;
                uint64_t e = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
                if (uint_as_float((uint32_t)(e >> 32)) < rec.t) { cur = (uint32_t)e; break; }
            }
        }
        if (cur == DONE) break;
    }
    if (rec.prim_idx < 0) rec.t = MAX_DIST;
    return rec;
}

// Any-hit occlusion over the 8-wide tree (unordered).
HD bool occlusion_test_bvh8(const BVH8Node* nodes,
                            const Prim* prims, const uint32_t* prim_obj,
                            const Ray& ray, float tmax,
                            uint64_t* lds_slot = nullptr, int lds_n = 0) {
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    int stack[BVH4_STACK];
    int sp = 0;
    int cur = 0;
    while (true) {
        const BVH8Node nd = nodes[cur];
        int next = -1;
#pragma unroll
        for (int c = 0; c < 8; ++c) {
            float t0x = fmaf(nd.lo_x[c], inv_d.x, -o_div.x);
            float t1x = fmaf(nd.hi_x[c], inv_d.x, -o_div.x);
            float t0y = fmaf(nd.lo_y[c], inv_d.y, -o_div.y);
            float t1y = fmaf(nd.hi_y[c], inv_d.y, -o_div.y);
            float t0z = fmaf(nd.lo_z[c], inv_d.z, -o_div.z);
            float t1z = fmaf(nd.hi_z[c], inv_d.z, -o_div.z);
            float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                fmaxf(fminf(t0z, t1z), 0.f));
            float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                fminf(fmaxf(t0z, t1z), tmax));
            if (enter > exit_) continue;
            int ch = nd.child[c];
            if (ch < 0) {
                int base = ~ch, pc = nd.cnt[c];
                for (int k = 0; k < pc; ++k) {
                    int pid = base + k;
                    bool sph = (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
                    float u, v;
                    float t = intersect_prim(prims[pid], sph, ray, u, v);
                    if (t > EPSILON && t < tmax) return true;
                }
            } else if (next < 0) {
                next = ch;
            } else {
                if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = (uint64_t)(uint32_t)ch;
                else stack[sp - lds_n] = ch;
                ++sp;
            }
        }
        if (next >= 0) { cur = next; continue; }
        if (sp == 0) return false;
        --sp;
        cur = sp < lds_n ? (int)(uint32_t)lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
    }
}

} // namespace hippt
