// bvh4.h — 4-wide BVH node layout + ordered short-stack traversal.
//
// Capability parity: same traversal semantics as the binary skip-link walk in
// bvh.h (reference src/renderer/tracing_func.cuh:44-181), but the node is
// re-designed for CDNA4 instead of copied: the binary walk is a serial
// dependent-load chain (measured VALUBusy ~4% on MI355X — the megakernel is
// latency-bound on L2/HBM, profiles/README.md), so the fix is to shorten the
// chain, not to widen the fetch of a single step (speculative dual-fetch was
// measured -18%).  A 4-wide node:
//   * is one 128-byte record = 8 independent 16-byte loads per step — the
//     memory parallelism lives INSIDE one step of the dependent chain,
//   * halves tree depth (half as many dependent steps per ray),
//   * tests 4 AABBs per step with pure VALU work, which is nearly free at 4%
//     VALU utilization,
//   * enables ordered (front-to-back) traversal with a short stack — the
//     DFS skip-link layout cannot reorder children, a 4-wide stack walk can,
//     and each stack entry carries t_near so stale far subtrees are culled
//     on pop after the hit distance tightens.
//
// Box storage is SoA-within-node (lo_x[4], lo_y[4], ... ) so each of the 4
// slab tests reads stride-16 floats from the same 2 cache lines.
#pragma once
#include "bvh.h"

namespace hippt {

// 128-byte 4-wide node.
//   child[c] >= 0 : internal child, index of a BVH4Node
//   child[c] <  0 : leaf child, prim_base = ~child[c], prim count = cnt[c]
//   cnt[c] == 0 AND child[c] == 0: empty slot (box is inverted: never hits)
struct alignas(16) BVH4Node {
    float lo_x[4], lo_y[4], lo_z[4];
    float hi_x[4], hi_y[4], hi_z[4];
    int32_t child[4];
    int32_t cnt[4];
};
static_assert(sizeof(BVH4Node) == 128, "BVH4Node must be 128 bytes");

constexpr int BVH4_STACK = 64;  // >= 3 * max collapsed depth; checked at build

// Intersect prims [base, base+cnt) and tighten rec.  tri_only (a scene-
// uniform scalar) skips the per-prim prim_obj sphere-bit load entirely —
// the reference's TRIANGLE_ONLY compile flag ("10% faster", defines.cuh:
// 26-27) as a runtime-free SGPR branch.
HD void bvh4_leaf_hit(const Prim* prims, const uint32_t* prim_obj,
                      const Ray& ray, int base, int cnt, HitRecord& rec,
                      bool tri_only = false) {
    for (int k = 0; k < cnt; ++k) {
        int pid = base + k;
        bool sph = !tri_only && (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
        float u, v;
        float t = intersect_prim(prims[pid], sph, ray, u, v);
        if (t > EPSILON && t < rec.t) {
            rec.t = t; rec.u = u; rec.v = v; rec.prim_idx = pid;
        }
    }
}

// Traversal stack: bottom lds_n entries live in LDS (device kernels pass a
// per-thread slot pointer into a __shared__ array, entry d at slot[d*256]
// for 256-thread blocks), the rest spill to a private (scratch) overflow
// array.  Scratch-backed stacks were measured to be the BVH4 bottleneck on
// MI355X (the walk is latency-bound; per-step 8-byte scratch push/pops
// thrash the vector caches), and the LDS allocation doubles as the
// occupancy governor: LDSN entries/thread x 8 B x 256 threads of LDS per
// block caps blocks/CU exactly where the scratch walk wants to run.
// On the host (lds_n = 0) everything goes to the plain array.
constexpr int BVH4_LDS_STRIDE = 256;  // all render blocks are 256 threads

// Closest-hit ordered traversal.
HD HitRecord ray_intersect_bvh4(const BVH4Node* nodes,
                                const Prim* prims, const uint32_t* prim_obj,
                                const Ray& ray, float tmax = MAX_DIST,
                                uint64_t* lds_slot = nullptr, int lds_n = 0) {
    HitRecord rec;
    rec.t = tmax;
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    // stack entry: (t_near bits << 32) | node index.  t_near >= 0 so the
    // float bit pattern orders like the float; on pop, entries whose t_near
    // is already beyond the current hit are skipped without a node load.
    uint64_t stack[BVH4_STACK];
    int sp = 0;
    int cur = 0;
    while (true) {
        const BVH4Node nd = nodes[cur];
        // 4 independent slab tests; keys sort hit children front-to-back.
        // key = (t_near bits & ~3) | slot — low 2 bits carry the slot index.
        uint32_t keys[4];
        int nhit = 0;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
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
            if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~3u) | (uint32_t)c;
        }
        // sort up to 4 keys ascending (nearest first): sorting network
        if (nhit > 1) {
            if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
            if (nhit > 2) {
                if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                if (nhit > 3) {
                    if (keys[2] > keys[3]) { uint32_t t = keys[2]; keys[2] = keys[3]; keys[3] = t; }
                    if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                    if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                }
            }
        }
        // near->far: leaves intersect inline (tightening rec.t before far
        // subtrees are entered); internals: nearest continues, rest pushed.
        int next = -1;
        for (int k = 0; k < nhit; ++k) {
            int c = (int)(keys[k] & 3u);
            int ch = nd.child[c];
            int pc = nd.cnt[c];
            if (ch < 0) {
                bvh4_leaf_hit(prims, prim_obj, ray, ~ch, pc, rec);
            } else if (next < 0) {
                next = ch;
            } else {
                uint64_t e = ((uint64_t)(keys[k] & ~3u) << 32) | (uint32_t)ch;
                if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = e;
                else stack[sp - lds_n] = e;
                ++sp;
            }
        }
        if (next >= 0) { cur = next; continue; }
        // pop, skipping subtrees now beyond the hit distance
        for (;;) {
            if (sp == 0) {
                if (rec.prim_idx < 0) rec.t = MAX_DIST;
                return rec;
            }
            --sp;
            uint64_t e = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
            if (uint_as_float((uint32_t)(e >> 32)) < rec.t) { cur = (int)(uint32_t)e; break; }
        }
    }
}

// While-while variant of the closest-hit walk (Aila & Laine style phase
// batching, re-derived for wave64): lanes alternate between a node-walk
// phase and a leaf-test phase at the OUTER loop level, so a wave tends to
// execute runs of same-kind work instead of interleaving a node step on
// some lanes with prim tests on others.  Leaf children are POSTPONED onto
// the stack (tagged entries) instead of intersected inline.  Measured A/B
// against the inline walk (divergence: VALUUtilization
// 15% on the inline walk).
//
// Stack/cur entry low word: bit31 = leaf, [30:27] = prim count (builders
// cap leaves at 15 prims), [26:0] = prim base (up to 134M prims).
HD HitRecord ray_intersect_bvh4_ww(const BVH4Node* nodes,
                                   const Prim* prims, const uint32_t* prim_obj,
                                   const Ray& ray, float tmax,
                                   uint64_t* lds_slot = nullptr, int lds_n = 0,
                                   const BVH4Node* top_cache = nullptr,
                                   int n_cached = 0, bool tri_only = false) {
    HitRecord rec;
    rec.t = tmax;
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    uint64_t stack[BVH4_STACK];
    int sp = 0;
    constexpr uint32_t DONE = 0x7fffffffu;
    uint32_t cur = 0;
    for (;;) {
        // ---- node phase: walk internal nodes until a leaf surfaces
        while (cur < 0x80000000u && cur != DONE) {
            // top-of-tree nodes come from the LDS copy (every walk starts at
            // the root; 95% L2 hit rate means the bound is hit LATENCY, and
            // LDS is ~4x closer than L2).  Pointer select, single flat load.
            const BVH4Node* nsrc = (int)cur < n_cached ? top_cache : nodes;
            const BVH4Node nd = nsrc[cur];
            uint32_t keys[4];
            int nhit = 0;
#pragma unroll
            for (int c = 0; c < 4; ++c) {
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
                if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~3u) | (uint32_t)c;
            }
            if (nhit > 1) {
                if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                if (nhit > 2) {
                    if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                    if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    if (nhit > 3) {
                        if (keys[2] > keys[3]) { uint32_t t = keys[2]; keys[2] = keys[3]; keys[3] = t; }
                        if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                        if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    }
                }
            }
            uint32_t next = DONE;
            for (int k = nhit - 1; k >= 0; --k) {  // far -> near so near pops first
                int c = (int)(keys[k] & 3u);
                int ch = nd.child[c];
                int pc = nd.cnt[c];
                if (ch < 0 && pc == 0) continue;   // empty slot
                uint32_t lo = ch < 0
                    ? (0x80000000u | ((uint32_t)pc << 27) | (uint32_t)(~ch))
                    : (uint32_t)ch;
                if (k == 0) {
                    next = lo;
                } else {
                    uint64_t e = ((uint64_t)(keys[k] & ~3u) << 32) | lo;
                    if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = e;
                    else stack[sp - lds_n] = e;
                    ++sp;
                }
            }
            if (next != DONE) { cur = next; continue; }
            for (;;) {  // pop, culling stale subtrees
                if (sp == 0) { cur = DONE; break; }
                --sp;
                uint64_t e = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
                if (uint_as_float((uint32_t)(e >> 32)) < rec.t) { cur = (uint32_t)e; break; }
            }
        }
        if (cur == DONE) break;
        // ---- leaf phase: drain consecutive leaf entries
        while (cur >= 0x80000000u) {
            bvh4_leaf_hit(prims, prim_obj, ray, (int)(cur & 0x07ffffffu),
                          (int)((cur >> 27) & 0xfu), rec, tri_only);
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

// Any-hit occlusion test: returns true if something blocks [EPSILON, tmax].
// No ordering (any hit ends the walk) — children are pushed unordered.
HD bool occlusion_test_bvh4(const BVH4Node* nodes,
                            const Prim* prims, const uint32_t* prim_obj,
                            const Ray& ray, float tmax,
                            uint64_t* lds_slot = nullptr, int lds_n = 0) {
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    int stack[BVH4_STACK];
    int sp = 0;
    int cur = 0;
    while (true) {
        const BVH4Node nd = nodes[cur];
        int next = -1;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
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

// Phase-batched any-hit occlusion walk: same while-while shaping as the
// closest-hit walk (leaf children postponed onto the stack so a wave runs
// node steps and prim tests in separate phases), but unordered and with an
// immediate `true` return on the first blocking hit.  A/B hook
// Measured +1-2% against the inline-leaf occlusion walk; now the default.
HD bool occlusion_test_bvh4_ww(const BVH4Node* nodes,
                               const Prim* prims, const uint32_t* prim_obj,
                               const Ray& ray, float tmax,
                               uint64_t* lds_slot = nullptr, int lds_n = 0,
                               const BVH4Node* top_cache = nullptr,
                               int n_cached = 0, bool tri_only = false) {
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    uint64_t stack[BVH4_STACK];
    int sp = 0;
    constexpr uint32_t DONE = 0x7fffffffu;
    uint32_t cur = 0;
    for (;;) {
        while (cur < 0x80000000u && cur != DONE) {
            const BVH4Node* nsrc = (int)cur < n_cached ? top_cache : nodes;
            const BVH4Node nd = nsrc[cur];
            uint32_t next = DONE;
#pragma unroll
            for (int c = 0; c < 4; ++c) {
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
                int pc = nd.cnt[c];
                if (ch < 0 && pc == 0) continue;
                uint32_t lo = ch < 0
                    ? (0x80000000u | ((uint32_t)pc << 27) | (uint32_t)(~ch))
                    : (uint32_t)ch;
                if (next == DONE) {
                    next = lo;
                } else {
                    uint64_t e = (uint64_t)lo;
                    if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = e;
                    else stack[sp - lds_n] = e;
                    ++sp;
                }
            }
            if (next != DONE) { cur = next; continue; }
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = (uint32_t)(sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE]
                                        : stack[sp - lds_n]);
        }
        if (cur == DONE) return false;
        while (cur >= 0x80000000u) {
            int base = (int)(cur & 0x07ffffffu);
            int pc = (int)((cur >> 27) & 0xfu);
            for (int k = 0; k < pc; ++k) {
                int pid = base + k;
                bool sph = (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
                float u, v;
                float t = intersect_prim(prims[pid], sph, ray, u, v);
                if (t > EPSILON && t < tmax) return true;
            }
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = (uint32_t)(sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE]
                                        : stack[sp - lds_n]);
        }
        if (cur == DONE) return false;
    }
}

// ----------------------------------------------------------------------
// Quantized 64-byte 4-wide node (Ylitie-style child-box compression,
// re-derived for this tree): child AABBs stored as uint8 offsets inside
// the node's own box, power-of-two per-axis scales so decompression is
// an fma per bound.  Quantized bounds round OUTWARD (conservative: never
// misses a hit, a few extra visits).  Halves bytes per dependent node
// load (1 cacheline, not 2) and doubles LDS top-cache coverage — the
// walk is latency-bound at 4% VALU, so the extra decompress math is free.
// Child words are pre-tagged exactly like traversal stack entries:
// bit31 leaf | [30:27] prim cnt | [26:0] base, or plain node index.
struct alignas(16) BVH4NodeQ {
    float ox, oy, oz;        // node box origin
    uint8_t ex, ey, ez;      // per-axis scale exponents: scale = 2^(e-135)
    uint8_t pad;
    uint8_t qlo[4][3];       // child lo offsets (floor)
    uint8_t qhi[4][3];       // child hi offsets (ceil); qhi<qlo = empty slot
    uint32_t child[4];       // pre-tagged child words (EMPTY_Q for empty)
};
static_assert(sizeof(BVH4NodeQ) == 64, "BVH4NodeQ must be 64 bytes");
constexpr uint32_t BVH4Q_EMPTY = 0x7fffffffu;

// scale = 2^(e-135): e chosen so extent/255 <= scale (exact for
// power-of-two extents, <2x conservative otherwise)
HD float bvh4q_scale(uint8_t e) {
    return uint_as_float((uint32_t)(e) << 23);  // 2^(e-127)
}

// Closest-hit while-while walk over the quantized tree.  Same phase
// batching, ordering and stack discipline as ray_intersect_bvh4_ww; child
// slab tests run on dequantized bounds.
HD HitRecord ray_intersect_bvh4q_ww(const BVH4NodeQ* nodes,
                                    const Prim* prims, const uint32_t* prim_obj,
                                    const Ray& ray, float tmax,
                                    uint64_t* lds_slot = nullptr, int lds_n = 0,
                                    const BVH4NodeQ* top_cache = nullptr,
                                    int n_cached = 0) {
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
            const BVH4NodeQ* nsrc = (int)cur < n_cached ? top_cache : nodes;
            const BVH4NodeQ nd = nsrc[cur];
            const float sx = bvh4q_scale(nd.ex), sy = bvh4q_scale(nd.ey), sz = bvh4q_scale(nd.ez);
            uint32_t keys[4];
            int nhit = 0;
#pragma unroll
            for (int c = 0; c < 4; ++c) {
                if (nd.child[c] == BVH4Q_EMPTY) continue;
                float lx = fmaf((float)nd.qlo[c][0], sx, nd.ox);
                float hx = fmaf((float)nd.qhi[c][0], sx, nd.ox);
                float ly = fmaf((float)nd.qlo[c][1], sy, nd.oy);
                float hy = fmaf((float)nd.qhi[c][1], sy, nd.oy);
                float lz = fmaf((float)nd.qlo[c][2], sz, nd.oz);
                float hz = fmaf((float)nd.qhi[c][2], sz, nd.oz);
                float t0x = fmaf(lx, inv_d.x, -o_div.x);
                float t1x = fmaf(hx, inv_d.x, -o_div.x);
                float t0y = fmaf(ly, inv_d.y, -o_div.y);
                float t1y = fmaf(hy, inv_d.y, -o_div.y);
                float t0z = fmaf(lz, inv_d.z, -o_div.z);
                float t1z = fmaf(hz, inv_d.z, -o_div.z);
                float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                    fmaxf(fminf(t0z, t1z), 0.f));
                float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                    fminf(fmaxf(t0z, t1z), rec.t));
                if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~3u) | (uint32_t)c;
            }
            if (nhit > 1) {
                if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                if (nhit > 2) {
                    if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                    if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    if (nhit > 3) {
                        if (keys[2] > keys[3]) { uint32_t t = keys[2]; keys[2] = keys[3]; keys[3] = t; }
                        if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                        if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    }
                }
            }
            uint32_t next = DONE;
            for (int k = nhit - 1; k >= 0; --k) {
                uint32_t lo = nd.child[keys[k] & 3u];
                if (k == 0) {
                    next = lo;
                } else {
                    uint64_t e = ((uint64_t)(keys[k] & ~3u) << 32) | lo;
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

// Any-hit occlusion walk over the quantized tree (unordered, early out).
HD bool occlusion_test_bvh4q_ww(const BVH4NodeQ* nodes,
                                const Prim* prims, const uint32_t* prim_obj,
                                const Ray& ray, float tmax,
                                uint64_t* lds_slot = nullptr, int lds_n = 0,
                                const BVH4NodeQ* top_cache = nullptr,
                                int n_cached = 0) {
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    uint64_t stack[BVH4_STACK];
    int sp = 0;
    constexpr uint32_t DONE = 0x7fffffffu;
    uint32_t cur = 0;
    for (;;) {
        while (cur < 0x80000000u && cur != DONE) {
            const BVH4NodeQ* nsrc = (int)cur < n_cached ? top_cache : nodes;
            const BVH4NodeQ nd = nsrc[cur];
            const float sx = bvh4q_scale(nd.ex), sy = bvh4q_scale(nd.ey), sz = bvh4q_scale(nd.ez);
            uint32_t next = DONE;
#pragma unroll
            for (int c = 0; c < 4; ++c) {
                if (nd.child[c] == BVH4Q_EMPTY) continue;
                float lx = fmaf((float)nd.qlo[c][0], sx, nd.ox);
                float hx = fmaf((float)nd.qhi[c][0], sx, nd.ox);
                float ly = fmaf((float)nd.qlo[c][1], sy, nd.oy);
                float hy = fmaf((float)nd.qhi[c][1], sy, nd.oy);
                float lz = fmaf((float)nd.qlo[c][2], sz, nd.oz);
                float hz = fmaf((float)nd.qhi[c][2], sz, nd.oz);
                float t0x = fmaf(lx, inv_d.x, -o_div.x);
                float t1x = fmaf(hx, inv_d.x, -o_div.x);
                float t0y = fmaf(ly, inv_d.y, -o_div.y);
                float t1y = fmaf(hy, inv_d.y, -o_div.y);
                float t0z = fmaf(lz, inv_d.z, -o_div.z);
                float t1z = fmaf(hz, inv_d.z, -o_div.z);
                float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                    fmaxf(fminf(t0z, t1z), 0.f));
                float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                    fminf(fmaxf(t0z, t1z), tmax));
                if (enter > exit_) continue;
                uint32_t lo = nd.child[c];
                if (next == DONE) {
                    next = lo;
                } else {
                    uint64_t e = (uint64_t)lo;
                    if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = e;
                    else stack[sp - lds_n] = e;
                    ++sp;
                }
            }
            if (next != DONE) { cur = next; continue; }
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = (uint32_t)(sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE]
                                        : stack[sp - lds_n]);
        }
        if (cur == DONE) return false;
        while (cur >= 0x80000000u) {
            int base = (int)(cur & 0x07ffffffu);
            int pc = (int)((cur >> 27) & 0xfu);
            for (int k = 0; k < pc; ++k) {
                int pid = base + k;
                bool sph = (prim_obj[pid] & PRIM_SPHERE_BIT) != 0;
                float u, v;
                float t = intersect_prim(prims[pid], sph, ray, u, v);
                if (t > EPSILON && t < tmax) return true;
            }
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = (uint32_t)(sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE]
                                        : stack[sp - lds_n]);
        }
        if (cur == DONE) return false;
    }
}

// Host-side conversion fp32 4-wide -> quantized (outward rounding).
inline std::vector<BVH4NodeQ> quantize_bvh4(const BVH4Node* nodes, int n) {
    std::vector<BVH4NodeQ> out((size_t)n);
    for (int i = 0; i < n; ++i) {
        const BVH4Node& s = nodes[i];
        BVH4NodeQ& q = out[i];
        // node box = union of child boxes
        float lo[3] = {3e38f, 3e38f, 3e38f}, hi[3] = {-3e38f, -3e38f, -3e38f};
        for (int c = 0; c < 4; ++c) {
            if (s.child[c] == 0 && s.cnt[c] == 0) continue;  // empty
            lo[0] = fminf(lo[0], s.lo_x[c]); hi[0] = fmaxf(hi[0], s.hi_x[c]);
            lo[1] = fminf(lo[1], s.lo_y[c]); hi[1] = fmaxf(hi[1], s.hi_y[c]);
            lo[2] = fminf(lo[2], s.lo_z[c]); hi[2] = fmaxf(hi[2], s.hi_z[c]);
        }
        if (lo[0] > hi[0]) { lo[0] = lo[1] = lo[2] = 0.f; hi[0] = hi[1] = hi[2] = 0.f; }
        q.ox = lo[0]; q.oy = lo[1]; q.oz = lo[2];
        uint8_t* eptr = &q.ex;
        for (int a = 0; a < 3; ++a) {
            float ext = fmaxf(hi[a] - lo[a], 0.f);
            // smallest power-of-two scale with 255*scale >= ext
            int e = 127;  // scale 1 -> covers ext <= 255
            if (ext > 0.f) {
                float need = ext / 255.f;
                int ee;
                frexpf(need, &ee);         // need = m * 2^ee, m in [0.5,1)
                e = 127 + ee;              // 2^ee >= need
                if (e < 1) e = 1;
                if (e > 254) e = 254;
            }
            eptr[a] = (uint8_t)e;
        }
        q.pad = 0;
        const float sxyz[3] = {bvh4q_scale(q.ex), bvh4q_scale(q.ey), bvh4q_scale(q.ez)};
        for (int c = 0; c < 4; ++c) {
            if (s.child[c] == 0 && s.cnt[c] == 0) {
                q.child[c] = BVH4Q_EMPTY;
                for (int a = 0; a < 3; ++a) { q.qlo[c][a] = 255; q.qhi[c][a] = 0; }
                continue;
            }
            const float clo[3] = {s.lo_x[c], s.lo_y[c], s.lo_z[c]};
            const float chi[3] = {s.hi_x[c], s.hi_y[c], s.hi_z[c]};
            for (int a = 0; a < 3; ++a) {
                float inv_s = 1.f / sxyz[a];
                float fl = floorf((clo[a] - (&q.ox)[a]) * inv_s);
                float fh = ceilf((chi[a] - (&q.ox)[a]) * inv_s);
                q.qlo[c][a] = (uint8_t)clampv((int)fl, 0, 255);
                q.qhi[c][a] = (uint8_t)clampv((int)fh, 0, 255);
            }
            q.child[c] = s.child[c] < 0
                ? (0x80000000u | ((uint32_t)s.cnt[c] << 27) | (uint32_t)(~s.child[c]))
                : (uint32_t)s.child[c];
        }
    }
    return out;
}

// Node-visit / prim-test counting walk for the BVH-cost visualizer
// (reference pt_impl/bvh_cost.cu:38-101; counts reflect the traversal that
// actually runs, i.e. the 4-wide one).
HD Vec2 bvh4_cost(const BVH4Node* nodes, const Prim* prims,
                  const uint32_t* prim_obj, const Ray& ray) {
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    float best_t = MAX_DIST;
    int node_visits = 0, prim_tests = 0;
    int stack[BVH4_STACK];
    int sp = 0;
    int cur = 0;
    while (true) {
        const BVH4Node nd = nodes[cur];
        ++node_visits;
        int next = -1;
        for (int c = 0; c < 4; ++c) {
            float t0x = fmaf(nd.lo_x[c], inv_d.x, -o_div.x);
            float t1x = fmaf(nd.hi_x[c], inv_d.x, -o_div.x);
            float t0y = fmaf(nd.lo_y[c], inv_d.y, -o_div.y);
            float t1y = fmaf(nd.hi_y[c], inv_d.y, -o_div.y);
            float t0z = fmaf(nd.lo_z[c], inv_d.z, -o_div.z);
            float t1z = fmaf(nd.hi_z[c], inv_d.z, -o_div.z);
            float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                fmaxf(fminf(t0z, t1z), 0.f));
            float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                fminf(fmaxf(t0z, t1z), best_t));
            if (enter > exit_) continue;
            int ch = nd.child[c];
            if (ch < 0) {
                int base = ~ch, pc = nd.cnt[c];
                for (int k = 0; k < pc; ++k) {
                    ++prim_tests;
                    float u, v;
                    bool sph = (prim_obj[base + k] & PRIM_SPHERE_BIT) != 0;
                    float t = intersect_prim(prims[base + k], sph, ray, u, v);
                    if (t > EPSILON && t < best_t) best_t = t;
                }
            } else if (next < 0) {
                next = ch;
            } else {
                stack[sp++] = ch;
            }
        }
        if (next >= 0) { cur = next; continue; }
        if (sp == 0) return {(float)node_visits, (float)prim_tests};
        cur = stack[--sp];
    }
}

// ----------------------------------------------------------------------
// Dual-ray while-while walk: one thread advances TWO independent closest-hit
// walks in lockstep steps, so each lane keeps two node loads in flight (the
// single-ray walk stalls ~58% on L2-hit latency; a second independent chain
// per lane doubles memory-level parallelism without needing more waves).
// Used by the wavefront trace kernel (HIPPT_WF_DUAL).  Each walk gets half
// of the thread's LDS stack slots.
struct Bvh4Walk {
    HitRecord rec;
    uint32_t cur;       // tagged entry: bit31 leaf, DONE_W = finished
    int sp;
    Vec3 inv_d, o_div;
    Ray ray;
    uint64_t stack[BVH4_STACK / 2];
};
constexpr uint32_t BVH4_DONE_W = 0x7fffffffu;

HD void bvh4_walk_init(Bvh4Walk& w, const Ray& ray, float tmax) {
    w.rec = HitRecord();
    w.rec.t = tmax;
    w.cur = 0;
    w.sp = 0;
    w.inv_d = safe_rcp_dir(ray.d);
    w.o_div = ray.o * w.inv_d;
    w.ray = ray;
}

// One step: a node visit (test 4 children, descend/push) OR one leaf batch.
// Returns true while the walk still has work.
HD bool bvh4_walk_step(Bvh4Walk& w, const BVH4Node* nodes, const Prim* prims,
                       const uint32_t* prim_obj, uint64_t* lds_slot, int lds_n) {
    if (w.cur == BVH4_DONE_W) return false;
    if (w.cur < 0x80000000u) {
        const BVH4Node nd = nodes[w.cur];
        uint32_t keys[4];
        int nhit = 0;
#pragma unroll
        for (int c = 0; c < 4; ++c) {
            float t0x = fmaf(nd.lo_x[c], w.inv_d.x, -w.o_div.x);
            float t1x = fmaf(nd.hi_x[c], w.inv_d.x, -w.o_div.x);
            float t0y = fmaf(nd.lo_y[c], w.inv_d.y, -w.o_div.y);
            float t1y = fmaf(nd.hi_y[c], w.inv_d.y, -w.o_div.y);
            float t0z = fmaf(nd.lo_z[c], w.inv_d.z, -w.o_div.z);
            float t1z = fmaf(nd.hi_z[c], w.inv_d.z, -w.o_div.z);
            float enter = fmaxf(fmaxf(fminf(t0x, t1x), fminf(t0y, t1y)),
                                fmaxf(fminf(t0z, t1z), 0.f));
            float exit_ = fminf(fminf(fmaxf(t0x, t1x), fmaxf(t0y, t1y)),
                                fminf(fmaxf(t0z, t1z), w.rec.t));
            if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~3u) | (uint32_t)c;
        }
        if (nhit > 1) {
            if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
            if (nhit > 2) {
                if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                if (nhit > 3) {
                    if (keys[2] > keys[3]) { uint32_t t = keys[2]; keys[2] = keys[3]; keys[3] = t; }
                    if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                    if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                }
            }
        }
        uint32_t next = BVH4_DONE_W;
        for (int k = nhit - 1; k >= 0; --k) {
            int c = (int)(keys[k] & 3u);
            int ch = nd.child[c];
            int pc = nd.cnt[c];
            if (ch < 0 && pc == 0) continue;
            uint32_t lo = ch < 0
                ? (0x80000000u | ((uint32_t)pc << 27) | (uint32_t)(~ch))
                : (uint32_t)ch;
            if (k == 0) {
                next = lo;
            } else {
                uint64_t e = ((uint64_t)(keys[k] & ~3u) << 32) | lo;
                if (w.sp < lds_n) lds_slot[w.sp * BVH4_LDS_STRIDE] = e;
                else w.stack[w.sp - lds_n] = e;
                ++w.sp;
            }
        }
        if (next != BVH4_DONE_W) { w.cur = next; return true; }
    } else {
        bvh4_leaf_hit(prims, prim_obj, w.ray, (int)(w.cur & 0x07ffffffu),
                      (int)((w.cur >> 27) & 0xfu), w.rec);
    }
    // pop (culling stale subtrees)
    for (;;) {
        if (w.sp == 0) { w.cur = BVH4_DONE_W; return false; }
        --w.sp;
        uint64_t e = w.sp < lds_n ? lds_slot[w.sp * BVH4_LDS_STRIDE] : w.stack[w.sp - lds_n];
        if (uint_as_float((uint32_t)(e >> 32)) < w.rec.t) { w.cur = (uint32_t)e; return true; }
    }
}

// 4-byte-entry variant of the while-while walk: stack entries carry only the
// tagged node/leaf word (no t_near, so no pop culling).  Halves stack LDS
// bytes -> twice the LDS entry capacity per thread at the same block budget.
// Measured 3% SLOWER than the 8-byte culling walk (host-tested record;
// no device path uses it).  lds_n here counts 4-byte entries.
HD HitRecord ray_intersect_bvh4_ww32(const BVH4Node* nodes,
                                     const Prim* prims, const uint32_t* prim_obj,
                                     const Ray& ray, float tmax,
                                     uint32_t* lds_slot = nullptr, int lds_n = 0) {
    HitRecord rec;
    rec.t = tmax;
    const Vec3 inv_d = safe_rcp_dir(ray.d);
    const Vec3 o_div = ray.o * inv_d;
    uint32_t stack[BVH4_STACK];
    int sp = 0;
    constexpr uint32_t DONE = 0x7fffffffu;
    uint32_t cur = 0;
    for (;;) {
        while (cur < 0x80000000u && cur != DONE) {
            const BVH4Node nd = nodes[cur];
            uint32_t keys[4];
            int nhit = 0;
#pragma unroll
            for (int c = 0; c < 4; ++c) {
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
                if (enter <= exit_) keys[nhit++] = (float_as_uint(enter) & ~3u) | (uint32_t)c;
            }
            if (nhit > 1) {
                if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                if (nhit > 2) {
                    if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                    if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    if (nhit > 3) {
                        if (keys[2] > keys[3]) { uint32_t t = keys[2]; keys[2] = keys[3]; keys[3] = t; }
                        if (keys[1] > keys[2]) { uint32_t t = keys[1]; keys[1] = keys[2]; keys[2] = t; }
                        if (keys[0] > keys[1]) { uint32_t t = keys[0]; keys[0] = keys[1]; keys[1] = t; }
                    }
                }
            }
            uint32_t next = DONE;
            for (int k = nhit - 1; k >= 0; --k) {
                int c = (int)(keys[k] & 3u);
                int ch = nd.child[c];
                int pc = nd.cnt[c];
                if (ch < 0 && pc == 0) continue;
                uint32_t lo = ch < 0
                    ? (0x80000000u | ((uint32_t)pc << 27) | (uint32_t)(~ch))
                    : (uint32_t)ch;
                if (k == 0) {
                    next = lo;
                } else {
                    if (sp < lds_n) lds_slot[sp * BVH4_LDS_STRIDE] = lo;
                    else stack[sp - lds_n] = lo;
                    ++sp;
                }
            }
            if (next != DONE) { cur = next; continue; }
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
        }
        if (cur == DONE) break;
        while (cur >= 0x80000000u) {
            bvh4_leaf_hit(prims, prim_obj, ray, (int)(cur & 0x07ffffffu),
                          (int)((cur >> 27) & 0xfu), rec);
            if (sp == 0) { cur = DONE; break; }
            --sp;
            cur = sp < lds_n ? lds_slot[sp * BVH4_LDS_STRIDE] : stack[sp - lds_n];
        }
        if (cur == DONE) break;
    }
    if (rec.prim_idx < 0) rec.t = MAX_DIST;
    return rec;
}

} // namespace hippt
