// wf_kernels.hip — wavefront path tracer for gfx950: SoA payload pool, fused
// shade/trace kernels, and a hand-written stable-enough 8-bit-key counting
// sort + stream compaction (replacing the reference's thrust::sort /
// partition / lower_bound pipeline).
//
// Capability parity: reference src/pt_impl/wavefront_pt.cu +
// wf_path_tracer.cu: raygen fused with primary closest hit (:71-133), fused
// ray-bounce shader (NEE + emitter-hit MIS + BSDF sampling, :213-314), fused
// closest-hit shader (next trace + miss/envmap + RR, :141-207), radiance
// splat (:477-503); index-buffer entries = 8-bit status {bit7 dead, low bits
// material id} over a 24-bit pixel id, sorted so same-material rays shade
// together and dead rays compact to the tail.
//
// MI355X-native design (round-2 final shape, every step A/B-measured —
// profiles/README.md):
//   * The sort is a 3-kernel one-byte counting sort (per-block LDS
//     histograms -> single-block scan -> scatter), memory-bound; the live
//     count is produced on-device and read back once per stage (4 bytes,
//     like the reference's post-sort lower_bound readback,
//     wf_path_tracer.cu:199).  The pipeline BREAKS at live==0, launches
//     live-sized grids, and gathers later sorts through the previous
//     compacted view so sort cost scales with the live count.
//   * Stages are FUSED: k_wf_primary / k_wf_step run `span` whole
//     shade+trace bounces in registers (single-source integrator
//     path_shade_hit/path_step, NEE occlusion inline) between payload
//     round trips.  The kitchen live-ray curve stays ~99% live through
//     bounce 6, so per-bounce sorting is pure overhead; the measured
//     optimum cadence is ~one sort/compaction per 8 bounces
//     (span1 124.5 -> span8 145.9 Msps, megakernel 149.5; and span8
//     BEATS span16-no-sort-at-all 144.3 — the compaction at ~40% live
//     measurably pays).
//   * Below HIPPT_WF_TAIL live rays (default 1.5M) one fused tail kernel
//     finishes every surviving path megakernel-style.
//   * HIPPT_WF_FUSE=0 / HIPPT_WF_SPAN=1 restore the classic split
//     shade / shadow-queue / trace per-bounce pipeline for A/B.
#include <hip/hip_runtime.h>
#include "kernels.h"
#include <cstdlib>
#include <cstring>
#include "../core/integrator.h"

namespace hippt {

constexpr int WF_BLOCK = 256;
constexpr int SORT_BLOCK = 256;
constexpr int SORT_ITEMS = 16;   // entries per thread in hist/scatter passes
constexpr uint32_t DEAD = 0x80u; // status bit 7

struct WfState {
    int w = 0, h = 0, n = 0;
    int nb_sort = 0;             // sort grid blocks
    float4* ray_o = nullptr;     // xyz + (unused)
    float4* ray_d = nullptr;     // xyz + prev_pdf
    float4* thp = nullptr;       // rgb + flags (bit0 prev_delta)
    float4* L = nullptr;         // rgb + lum2 accumulation for variance
    float4* hit = nullptr;       // t, u, v, prim_idx (int bits)
    float4* prevn = nullptr;     // previous shading normal + bounce counters packed in w
    unsigned long long* rng = nullptr;
    uint32_t* status = nullptr;  // (status<<24)|payload, PIXEL order (trace input)
    uint32_t* order = nullptr;   // compacted material-sorted view (shade input)
    uint32_t* order2 = nullptr;  // double buffer: bounce b gathers via b-1's view
    uint32_t* hist = nullptr;    // nb_sort * 256
    int* live_dev = nullptr;
    int* live_host = nullptr;    // pinned
    // shadow-ray queue: NEE visibility moved out of the fat shade kernel into
    // a lean 8-wave-per-SIMD traversal kernel (MI355X occupancy design)
    float4* sh_od = nullptr;     // origin xyz + tmax
    float4* sh_dir = nullptr;    // dir xyz + payload idx (uint bits)
    float4* sh_val = nullptr;    // premultiplied contribution rgb
    int* sh_cnt = nullptr;       // device append counter
};

// lobe counters packed into prevn.w: 4 x 8-bit (diffuse, specular, transmit, total)
__device__ __forceinline__ uint32_t pack_counts(int d, int s, int t, int b) {
    return (uint32_t)(d | (s << 8) | (t << 16) | (b << 24));
}

// Dynamic-LDS layout for all traversal kernels here, same as the megakernel
// (pt_kernels.hip k_render): [n_cached 128-byte top-tree nodes][stacks].
// Every walk's first visits are nodes 0..n_cached and the walk is bound by
// hit LATENCY (95% L2 hit rate) — LDS is ~4x closer than L2.  Round-1 WFPT
// omitted this cache; it was the bulk of the megakernel-vs-wavefront gap
// (trace alone cost as much as the whole megakernel, profiles/README.md r02).
__device__ __forceinline__ TravCtx wf_lds_ctx(const SceneView& sv, uint64_t* s_stk,
                                              int lds_n, int n_cached) {
    constexpr int NW = (int)(sizeof(TravNode) / 8);   // u64 words per node
    TravNode* s_cache = (TravNode*)s_stk;
    uint64_t* s_base = s_stk + (size_t)n_cached * NW;
    for (int i = threadIdx.x; i < n_cached * NW; i += 256)
        ((uint64_t*)s_cache)[i] = ((const uint64_t*)trav_nodes(sv))[i];
    if (n_cached > 0) __syncthreads();
    return TravCtx{&s_base[threadIdx.x], lds_n, s_cache, n_cached};
}

// ----------------------------------------------------------------- raygen
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))  // exact residency = what the
// LDS stack budget allows; frees 512/MINW VGPRs so the walk state does not
// spill (same fix as the megakernel, measured +9% there)
void k_wf_raygen(SceneView sv, WfState st, int spp_idx, uint32_t seed, int lds_n,
                 int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= st.n) return;
    int px = i % st.w, py = i / st.w;
    Sampler sp(uint32_t(i), uint32_t(spp_idx) * SEED_SCALER + seed);
    Ray ray = sv.cam.gen_ray(px, py, sp, spp_idx);
    HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
    st.ray_o[i] = make_float4(ray.o.x, ray.o.y, ray.o.z, 0.f);
    st.ray_d[i] = make_float4(ray.d.x, ray.d.y, ray.d.z, 0.f);
    st.thp[i] = make_float4(1.f, 1.f, 1.f, uint_as_float(1u));  // prev_delta=1
    st.L[i] = make_float4(0.f, 0.f, 0.f, 0.f);
    st.hit[i] = make_float4(hit.t, hit.u, hit.v, int_as_float(hit.prim_idx));
    st.prevn[i] = make_float4(0.f, 0.f, 1.f, uint_as_float(pack_counts(0, 0, 0, 0)));
    st.rng[i] = sp.state;
    uint32_t status;
    if (hit.prim_idx < 0) {
        status = DEAD;  // miss: envmap resolved in the first shade pass? no -> here
        if (sv.env_emitter >= 0) {
            Vec3 le = envmap_eval(sv.emitters[sv.env_emitter], ray.d, sv.textures);
            st.L[i] = make_float4(le.x, le.y, le.z, 0.f);
        }
    } else {
        uint32_t oi = sv.prim_obj[hit.prim_idx] & PRIM_OBJ_MASK;
        status = (uint32_t)(sv.objs[oi].bsdf_id & 0x3F);
    }
    st.status[i] = (status << 24) | (uint32_t)i;
}

// ------------------------------------------------------------ counting sort
// Grouping sort (order within a bin is arrival order, which preserves
// block-local pixel adjacency): one global 256-bin histogram, a 1-block
// exclusive scan producing the live count on-device, and an atomic-cursor
// scatter.  No host round trip; 3 memory-bound passes over 4B/entry.
// mode 0: key = material byte (grouped shade).  mode 1: key = dead bit only
// (compaction without ray sorting — reference NO_RAY_SORTING/partition path,
// defines.cuh:25-37; keeps arrival (pixel) order within live/dead).
__device__ inline uint32_t sort_key(uint32_t status, int mode) {
    uint32_t b = status >> 24;
    return mode ? (b >= DEAD ? (uint32_t)DEAD : 0u) : b;
}

// `gather` non-null: the pass runs over the previous bounce's compacted
// order view (n = prev live count) and re-reads each entry's FRESH status
// byte from the pixel-order status array — so from bounce 1 on, sort cost
// scales with the live count, not W*H.
__device__ __forceinline__ uint32_t sort_load(const uint32_t* in, const uint32_t* gather, int i) {
    return gather ? in[gather[i] & 0x00FFFFFFu] : in[i];
}

__global__ __launch_bounds__(SORT_BLOCK)
void k_sort_hist(const uint32_t* __restrict__ in, const uint32_t* __restrict__ gather,
                 int n, uint32_t* __restrict__ hist, int mode) {
    __shared__ uint32_t lh[256];
    for (int t = threadIdx.x; t < 256; t += blockDim.x) lh[t] = 0;
    __syncthreads();
    int base = blockIdx.x * SORT_BLOCK * SORT_ITEMS;
    for (int k = 0; k < SORT_ITEMS; ++k) {
        int i = base + k * SORT_BLOCK + threadIdx.x;
        if (i < n) atomicAdd(&lh[sort_key(sort_load(in, gather, i), mode)], 1u);
    }
    __syncthreads();
    for (int t = threadIdx.x; t < 256; t += blockDim.x)
        if (lh[t]) atomicAdd(&hist[t], lh[t]);
}

// single block of 256: exclusive scan of the 256 bins -> cursors + live count
__global__ __launch_bounds__(256)
void k_sort_scan(uint32_t* __restrict__ hist, int* __restrict__ live_out) {
    __shared__ uint32_t base[256];
    int b = threadIdx.x;
    uint32_t x = hist[b];
    base[b] = x;
    __syncthreads();
    for (int off = 1; off < 256; off <<= 1) {
        uint32_t v = (b >= off) ? base[b - off] : 0;
        __syncthreads();
        base[b] += v;
        __syncthreads();
    }
    uint32_t excl = base[b] - x;
    hist[b] = excl;          // becomes the atomic cursor for scatter
    if (b == (int)DEAD - 1 && live_out) *live_out = (int)base[b];
}

__global__ __launch_bounds__(SORT_BLOCK)
void k_sort_scatter(const uint32_t* __restrict__ in, const uint32_t* __restrict__ gather,
                    int n, uint32_t* __restrict__ hist, uint32_t* __restrict__ out, int mode) {
    __shared__ uint32_t lbase[256];
    __shared__ uint32_t lcnt[256];
    for (int t = threadIdx.x; t < 256; t += blockDim.x) lcnt[t] = 0;
    __syncthreads();
    // block-local ranks first, then one global cursor bump per (block, bin)
    int base = blockIdx.x * SORT_BLOCK * SORT_ITEMS;
    uint32_t rank[SORT_ITEMS];
    uint32_t ent[SORT_ITEMS];
    for (int k = 0; k < SORT_ITEMS; ++k) {
        int i = base + k * SORT_BLOCK + threadIdx.x;
        if (i < n) {
            ent[k] = sort_load(in, gather, i);
            rank[k] = atomicAdd(&lcnt[sort_key(ent[k], mode)], 1u);
        }
    }
    __syncthreads();
    for (int t = threadIdx.x; t < 256; t += blockDim.x)
        lbase[t] = lcnt[t] ? atomicAdd(&hist[t], lcnt[t]) : 0;
    __syncthreads();
    // dead entries are dropped (nothing downstream reads past the live
    // prefix), so the compacted view holds live rays only
    for (int k = 0; k < SORT_ITEMS; ++k) {
        int i = base + k * SORT_BLOCK + threadIdx.x;
        if (i < n) {
            uint32_t key = sort_key(ent[k], mode);
            if (key < DEAD) out[lbase[key] + rank[k]] = ent[k];
        }
    }
}

// ----------------------------------------------------------- bounce shade
// NEE + emitter-hit MIS + BSDF sample for live rays (current hit record).
__global__ __launch_bounds__(256)
void k_wf_shade(SceneView sv, WfState st, const uint32_t* __restrict__ order,
                int live, int bounce) {
    int k = blockIdx.x * blockDim.x + threadIdx.x;
    if (k >= live) return;
    uint32_t entry = order[k];
    int i = (int)(entry & 0x00FFFFFFu);
    float4 h4 = st.hit[i];
    int prim_idx = float_as_int(h4.w);
    // (live entries always have a valid hit)
    float4 ro4 = st.ray_o[i], rd4 = st.ray_d[i], thp4 = st.thp[i], l4 = st.L[i];
    Ray ray(Vec3(ro4.x, ro4.y, ro4.z), Vec3(rd4.x, rd4.y, rd4.z));
    Vec3 thp(thp4.x, thp4.y, thp4.z), L(l4.x, l4.y, l4.z);
    float prev_pdf = rd4.w;
    bool prev_delta = (float_as_uint(thp4.w) & 1u) != 0;
    float4 pn4 = st.prevn[i];
    Vec3 prev_n(pn4.x, pn4.y, pn4.z);
    uint32_t counts = float_as_uint(pn4.w);
    Sampler sp(st.rng[i]);

    Vec3 pos = ray.at(h4.x);
    uint32_t po = sv.prim_obj[prim_idx];
    bool is_sphere = (po & PRIM_SPHERE_BIT) != 0;
    const ObjInfo& obj = sv.objs[po & PRIM_OBJ_MASK];
    const Prim prim = sv.prims[prim_idx];
    Interaction it = get_interaction(prim, sv.attrs[prim_idx], is_sphere, pos, h4.y, h4.z);
    const BsdfParams& bsdf = sv.bsdfs[obj.bsdf_id];
    if (bsdf.tex[TEX_NORMAL] >= 0)
        it.shading_n = apply_normal_map(sv.textures, bsdf.tex[TEX_NORMAL], it.uv, it.shading_n);

    // emitter-hit MIS accumulation
    if (obj.emitter_id >= 0) {
        const EmitterParams& em = sv.emitters[obj.emitter_id];
        Vec3 le = emitter_eval_le(em, it.shading_n, -ray.d, it.uv, sv.textures);
        if (!le.is_zero()) {
            float w = 1.f;
            if (!prev_delta) {
                float light_pdf = emitter_pdf_hit(em, ray.d, h4.x, it.shading_n, prev_n, sv.emitter_geom()) *
                                emitter_sel_pdf(sv, obj.emitter_id);
                w = mis_weight(prev_pdf, light_pdf);
            }
            L += thp * le * w;
        }
    }

    // NEE: enqueue the shadow ray; visibility resolved by k_wf_shadow
    if (!bsdf_is_delta(bsdf) && sv.n_emitters > 0) {
        float epdf;
        int ei = pick_emitter(sv, sp, epdf);
        EmitterSampleRec er = emitter_sample(sv.emitters[ei], sv.emitter_geom(), pos,
                                             it.shading_n, sp);
        if (er.pdf > 0.f && !er.radiance.is_zero()) {
            Vec3 to_l = er.pos - pos;
            float dist = to_l.length();
            Vec3 wi = to_l * (1.f / fmaxf(dist, 1e-9f));
            Vec3 f = bsdf_eval(bsdf, -ray.d, wi, it, sv.textures);
            if (!f.is_zero()) {
                float sh_max = (sv.emitters[ei].type == EM_ENVMAP ? ENVMAP_DIST : dist) - 2.f * EPSILON;
                float light_pdf = er.pdf * epdf;
                float w = er.delta ? 1.f
                                   : mis_weight(light_pdf, bsdf_pdf(bsdf, -ray.d, wi, it, sv.textures));
                Vec3 val = thp * f * er.radiance * (w / light_pdf);
                if (!val.is_zero() && !val.has_nan()) {
                    int slot = atomicAdd(st.sh_cnt, 1);
                    Vec3 so = fmadd(wi, EPSILON, pos);
                    st.sh_od[slot] = make_float4(so.x, so.y, so.z, sh_max);
                    st.sh_dir[slot] = make_float4(wi.x, wi.y, wi.z, uint_as_float((uint32_t)i));
                    st.sh_val[slot] = make_float4(val.x, val.y, val.z, 0.f);
                }
            }
        }
    }

    // BSDF sample (per-path dispersion wavelength lives in ray_o.w)
    uint32_t status = DEAD;
    float path_lambda = ro4.w;
    BsdfSample bs = bsdf_sample(bsdf, -ray.d, it, sp, sv.textures, &path_lambda);
    if (bs.pdf > 0.f && !bs.weight.is_zero() && !bs.weight.has_nan() && !bs.wi.has_nan()) {
        thp *= bs.weight;
        int nd = counts & 0xFF, ns = (counts >> 8) & 0xFF, nt = (counts >> 16) & 0xFF,
            nb = (counts >> 24) & 0xFF;
        bool over = false;
        if (!(bs.lobe & LOBE_NULL)) {
            if (bs.lobe & LOBE_DIFFUSE)  over |= (++nd > sv.md.max_diffuse);
            if (bs.lobe & LOBE_SPECULAR) over |= (++ns > sv.md.max_specular);
            if (bs.lobe & LOBE_TRANSMIT) over |= (++nt > sv.md.max_transmit);
            ++nb;
        }
        over |= (nb >= sv.md.max_depth);
        // Russian roulette (threshold 0.1 after bounce 1)
        if (!over && nb > 1) {
            float p = clampv(thp.max_elem(), 0.f, 1.f);
            if (p < 0.1f) {
                if (sp.next1f() >= p * 10.f) over = true;
                else thp *= (1.f / (p * 10.f));
            }
        }
        if (!over) {
            status = 0;  // alive; material filled by the trace kernel
            counts = pack_counts(nd, ns, nt, nb);
            prev_delta = (bs.lobe & LOBE_DELTA) != 0;
            prev_pdf = bs.pdf;
            prev_n = it.shading_n;
            Vec3 no = fmadd(bs.wi, EPSILON, pos);
            st.ray_o[i] = make_float4(no.x, no.y, no.z, path_lambda);
            st.ray_d[i] = make_float4(bs.wi.x, bs.wi.y, bs.wi.z, prev_pdf);
        }
    }
    st.thp[i] = make_float4(thp.x, thp.y, thp.z, uint_as_float(prev_delta ? 1u : 0u));
    st.L[i] = make_float4(L.x, L.y, L.z, l4.w);
    st.prevn[i] = make_float4(prev_n.x, prev_n.y, prev_n.z, uint_as_float(counts));
    st.rng[i] = sp.state;
    // write status at the PAYLOAD index (status array stays in pixel order;
    // the trace kernel scans it coherently); material id refined by trace
    st.status[i] = (status << 24) | (uint32_t)i;
}

// ---------------------------------------------------- shadow-ray resolve
// Lean traversal-only kernel (58 VGPR class -> 8 waves/SIMD): any-hit test,
// then a race-free add into L (exactly one shadow ray per payload per bounce).
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))  // exact residency = what the
// LDS stack budget allows; frees 512/MINW VGPRs so the walk state does not
// spill (same fix as the megakernel, measured +9% there)
void k_wf_shadow(SceneView sv, WfState st, int lds_n, int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int k = blockIdx.x * blockDim.x + threadIdx.x;
    if (k >= *st.sh_cnt) return;
    float4 od = st.sh_od[k];
    float4 dir = st.sh_dir[k];
    Ray ray(Vec3(od.x, od.y, od.z), Vec3(dir.x, dir.y, dir.z));
    if (scene_occluded(sv, ray, od.w, tc)) return;
    int i = (int)float_as_uint(dir.w);
    float4 v = st.sh_val[k];
    float4 l4 = st.L[i];
    st.L[i] = make_float4(l4.x + v.x, l4.y + v.y, l4.z + v.z, l4.w);
}

// ------------------------------------------------------- next closest hit
// trace epilogue shared by the single- and dual-ray trace kernels: store
// the hit record, accumulate envmap MIS on miss, refine the status byte.
__device__ inline void wf_trace_finish(const SceneView& sv, WfState& st, int i,
                                       const Ray& ray, float prev_pdf,
                                       HitRecord hit) {
    if (hit.prim_idx < 0) hit.t = MAX_DIST;
    st.hit[i] = make_float4(hit.t, hit.u, hit.v, int_as_float(hit.prim_idx));
    uint32_t status;
    if (hit.prim_idx < 0) {
        status = DEAD;
        if (sv.env_emitter >= 0) {
            float4 thp4 = st.thp[i];
            float4 pn4 = st.prevn[i];
            bool prev_delta = (float_as_uint(thp4.w) & 1u) != 0;
            const EmitterParams& env = sv.emitters[sv.env_emitter];
            Vec3 le = envmap_eval(env, ray.d, sv.textures);
            float w = 1.f;
            if (!prev_delta) {
                float light_pdf = emitter_pdf_hit(env, ray.d, ENVMAP_DIST, ray.d,
                                                  Vec3(pn4.x, pn4.y, pn4.z),
                                                  sv.emitter_geom()) *
                                  emitter_sel_pdf(sv, sv.env_emitter);
                w = mis_weight(prev_pdf, light_pdf);
            }
            float4 l4 = st.L[i];
            Vec3 L = Vec3(l4.x, l4.y, l4.z) + Vec3(thp4.x, thp4.y, thp4.z) * le * w;
            st.L[i] = make_float4(L.x, L.y, L.z, l4.w);
        }
    } else {
        uint32_t oi = sv.prim_obj[hit.prim_idx] & PRIM_OBJ_MASK;
        status = (uint32_t)(sv.objs[oi].bsdf_id & 0x3F);
    }
    st.status[i] = (status << 24) | (uint32_t)i;
}

// Live-prefix trace: scans the compacted order view (grid sized from the
// live count).  Entries within a material bin keep block-local pixel
// adjacency from the scatter, so wave-level spatial coherence survives the
// material grouping.  Shade may have terminated a ray after the sort ran,
// so the fresh status byte is still checked.
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))  // exact residency = what the
// LDS stack budget allows; frees 512/MINW VGPRs so the walk state does not
// spill (same fix as the megakernel, measured +9% there)
void k_wf_trace(SceneView sv, WfState st, const uint32_t* __restrict__ order,
                int live, int lds_n, int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int k = blockIdx.x * blockDim.x + threadIdx.x;
    if (k >= live) return;
    int i = (int)(order[k] & 0x00FFFFFFu);
    if (st.status[i] >> 24 >= DEAD) return;  // terminated in shade
    float4 ro4 = st.ray_o[i], rd4 = st.ray_d[i];
    Ray ray(Vec3(ro4.x, ro4.y, ro4.z), Vec3(rd4.x, rd4.y, rd4.z));
    HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
    wf_trace_finish(sv, st, i, ray, rd4.w, hit);
}

// Dual-ray trace: each lane advances two independent walks in lockstep
// steps, keeping two node loads in flight per lane (the single walk stalls
// ~58% of cycles on L2-hit latency).  Order entries k and k + live/2 pair
// up so both halves stay in compaction order.  HIPPT_WF_DUAL selects this
// kernel; the launcher refuses it unless 3*bvh4_depth fits the halved
// stack (BVH4_STACK/2 private + lds_n/2 LDS entries per walk).
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(4, 4)))   // 2x walk state needs registers
void k_wf_trace_dual(SceneView sv, WfState st, const uint32_t* __restrict__ order,
                     int live, int lds_n) {
    extern __shared__ uint64_t s_stk[];
    const int tid = threadIdx.x;
    const int half = (live + 1) >> 1;
    const int k0 = blockIdx.x * blockDim.x + tid;
    const int k1 = k0 + half;
    const int lds_half = lds_n >> 1;
    uint64_t* slot0 = &s_stk[tid];
    uint64_t* slot1 = &s_stk[tid + (size_t)lds_half * BVH4_LDS_STRIDE];
    const int i0 = k0 < half ? (int)(order[k0] & 0x00FFFFFFu) : 0;
    const int i1 = k1 < live ? (int)(order[k1] & 0x00FFFFFFu) : 0;
    bool a0 = k0 < half && (st.status[i0] >> 24) < DEAD;
    bool a1 = k1 < live && (st.status[i1] >> 24) < DEAD;
    if (!a0 && !a1) return;
    Ray ray0, ray1;
    float pdf0 = 0.f, pdf1 = 0.f;
    Bvh4Walk w0, w1;
    if (a0) {
        float4 ro = st.ray_o[i0], rd = st.ray_d[i0];
        ray0 = Ray(Vec3(ro.x, ro.y, ro.z), Vec3(rd.x, rd.y, rd.z));
        pdf0 = rd.w;
        bvh4_walk_init(w0, ray0, MAX_DIST);
    }
    if (a1) {
        float4 ro = st.ray_o[i1], rd = st.ray_d[i1];
        ray1 = Ray(Vec3(ro.x, ro.y, ro.z), Vec3(rd.x, rd.y, rd.z));
        pdf1 = rd.w;
        bvh4_walk_init(w1, ray1, MAX_DIST);
    }
    bool r0 = a0, r1 = a1;
    while (r0 | r1) {
        if (r0) r0 = bvh4_walk_step(w0, sv.nodes4, sv.prims, sv.prim_obj, slot0, lds_half);
        if (r1) r1 = bvh4_walk_step(w1, sv.nodes4, sv.prims, sv.prim_obj, slot1, lds_half);
    }
    if (a0) wf_trace_finish(sv, st, i0, ray0, pdf0, w0.rec);
    if (a1) wf_trace_finish(sv, st, i1, ray1, pdf1, w1.rec);
}

// --------------------------------------------- payload <-> PathState glue
// The wavefront payload pool resumes paths through the SAME single-source
// integrator as the megakernel (path_shade_hit / path_step).
__device__ __forceinline__ void wf_load_ps(const WfState& st, int i, PathState& ps,
                                           Sampler& sp, HitRecord& hit) {
    float4 ro4 = st.ray_o[i], rd4 = st.ray_d[i], thp4 = st.thp[i], l4 = st.L[i];
    float4 h4 = st.hit[i], pn4 = st.prevn[i];
    ps.ray = Ray(Vec3(ro4.x, ro4.y, ro4.z), Vec3(rd4.x, rd4.y, rd4.z));
    ps.L = Vec3(l4.x, l4.y, l4.z);
    ps.thp = Vec3(thp4.x, thp4.y, thp4.z);
    ps.prev_pdf = rd4.w;
    ps.prev_delta = (float_as_uint(thp4.w) & 1u) != 0;
    ps.prev_n = Vec3(pn4.x, pn4.y, pn4.z);
    ps.path_time = 0.f;
    uint32_t counts = float_as_uint(pn4.w);
    ps.st.n_diffuse = (int)(counts & 0xFF);
    ps.st.n_specular = (int)((counts >> 8) & 0xFF);
    ps.st.n_transmit = (int)((counts >> 16) & 0xFF);
    ps.st.n_volume = 0;
    ps.b = (int)((counts >> 24) & 0xFF);
    ps.iter = ps.b + 1;  // null bounces before the handoff are not replayed
    ps.aov_n = Vec3(0.f); ps.aov_alb = Vec3(0.f); ps.aov_t = 0.f;
    ps.lambda = ro4.w;
    sp = Sampler(st.rng[i]);
    hit.t = h4.x; hit.u = h4.y; hit.v = h4.z;
    hit.prim_idx = float_as_int(h4.w);
}

__device__ __forceinline__ void wf_store_ps(WfState& st, int i, const PathState& ps,
                                            const Sampler& sp) {
    st.ray_o[i] = make_float4(ps.ray.o.x, ps.ray.o.y, ps.ray.o.z, ps.lambda);
    st.ray_d[i] = make_float4(ps.ray.d.x, ps.ray.d.y, ps.ray.d.z, ps.prev_pdf);
    st.thp[i] = make_float4(ps.thp.x, ps.thp.y, ps.thp.z,
                            uint_as_float(ps.prev_delta ? 1u : 0u));
    st.L[i] = make_float4(ps.L.x, ps.L.y, ps.L.z, 0.f);
    st.prevn[i] = make_float4(ps.prev_n.x, ps.prev_n.y, ps.prev_n.z,
                              uint_as_float(pack_counts(ps.st.n_diffuse, ps.st.n_specular,
                                                        ps.st.n_transmit, ps.b)));
    st.rng[i] = sp.state;
}

// Shared shade(+trace) core of the fused step/primary kernels: shade the
// in-register hit, trace the sampled ray, store payload + status.
__device__ __forceinline__ void wf_step_core(const SceneView& sv, WfState& st, int i,
                                             PathState& ps, Sampler& sp, TravCtx tc,
                                             const HitRecord& hit, int extra_bounces,
                                             int do_trace) {
    bool done = path_shade_hit(sv, ps, sp, tc, hit);
    // multi-bounce span: keep shading+tracing in registers while live
    // fractions are high (kitchen stays ~99% live through bounce 6), so
    // the payload round trip + sort only happen every `span` bounces
    for (int b = 0; b < extra_bounces && !done; ++b)
        done = path_step(sv, ps, sp, tc);
    if (!done && ps.b >= sv.md.max_depth) done = true;  // path_step entry cap
    uint32_t status = DEAD;
    if (!done && do_trace) {
        HitRecord nh = scene_intersect(sv, ps.ray, MAX_DIST, tc);
        if (nh.prim_idx < 0) {
            nh.t = MAX_DIST;
            path_shade_hit(sv, ps, sp, tc, nh);  // miss branch: envmap + MIS
        } else {
            uint32_t oi = sv.prim_obj[nh.prim_idx] & PRIM_OBJ_MASK;
            status = (uint32_t)(sv.objs[oi].bsdf_id & 0x3F);
        }
        st.hit[i] = make_float4(nh.t, nh.u, nh.v, int_as_float(nh.prim_idx));
    }
    wf_store_ps(st, i, ps, sp);
    st.status[i] = (status << 24) | (uint32_t)i;
}

// ------------------------------------------------------- fused bounce step
// One kernel per bounce: shade the stored hit (NEE occlusion inline — the
// megakernel-measured winner at occ 4) and immediately trace the sampled
// ray while it is still in registers.  Versus the split
// shade/shadow-queue/trace pipeline this removes two kernel boundaries
// worth of payload HBM round trips per bounce and gives the latency-bound
// BVH walk shade VALU work to overlap.  The sort/compaction stays — the
// wavefront's actual value on this hardware.  HIPPT_WF_FUSE=0 restores the
// split pipeline for A/B.
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))
void k_wf_step(SceneView sv, WfState st, const uint32_t* __restrict__ order,
               int live, int extra_bounces, int do_trace, int lds_n, int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int k = blockIdx.x * blockDim.x + threadIdx.x;
    if (k >= live) return;
    int i = (int)(order[k] & 0x00FFFFFFu);
    PathState ps;
    Sampler sp(0, 0);
    HitRecord hit;
    wf_load_ps(st, i, ps, sp, hit);
    wf_step_core(sv, st, i, ps, sp, tc, hit, extra_bounces, do_trace);
}

// ------------------------------------------------------ fused primary step
// Camera ray + primary hit + bounce-0 shade + bounce-1 trace in ONE kernel
// (replaces the separate raygen: no payload round trip or sort before the
// first shade — primary rays are pixel-coherent anyway, and the
// material-grouped shade was measured neutral).
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))
void k_wf_primary(SceneView sv, WfState st, int spp_idx, uint32_t seed,
                  int extra_bounces, int do_trace, int lds_n, int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= st.n) return;
    int px = i % st.w, py = i / st.w;
    Sampler sp(uint32_t(i), uint32_t(spp_idx) * SEED_SCALER + seed);
    Ray ray = sv.cam.gen_ray(px, py, sp, spp_idx);
    PathState ps;
    ps.reset(ray);
    ps.iter = 1;
    HitRecord hit = scene_intersect(sv, ray, MAX_DIST, tc);
    if (hit.prim_idx < 0) hit.t = MAX_DIST;
    wf_step_core(sv, st, i, ps, sp, tc, hit, extra_bounces, do_trace);
}

// ------------------------------------------------------------- tail fuse
// Below HIPPT_WF_TAIL live rays the chip is under-occupied anyway, so
// per-bounce sorting + kernel launches cost more than the wave64
// divergence they avoid.  This kernel finishes every surviving path
// megakernel-style: shade the stored hit (integrator.h path_shade_hit),
// then loop full path_step bounces (inline NEE occlusion, no shadow
// queue) until the path dies.  Mirrors the reference's early loop exit at
// live == 0 (wf_path_tracer.cu:199-210) taken one step further.
template <int MINW = 4>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))  // exact residency = what the
// LDS stack budget allows; frees 512/MINW VGPRs so the walk state does not
// spill (same fix as the megakernel, measured +9% there)
void k_wf_tail(SceneView sv, WfState st, const uint32_t* __restrict__ order,
               int live, int lds_n, int n_cached) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc = wf_lds_ctx(sv, s_stk, lds_n, n_cached);
    int k = blockIdx.x * blockDim.x + threadIdx.x;
    if (k >= live) return;
    int i = (int)(order[k] & 0x00FFFFFFu);
    PathState ps;
    Sampler sp(0, 0);
    HitRecord hit;
    wf_load_ps(st, i, ps, sp, hit);
    bool done = path_shade_hit(sv, ps, sp, tc, hit);
    while (!done) done = path_step(sv, ps, sp, tc);
    float lw = st.L[i].w;
    st.L[i] = make_float4(ps.L.x, ps.L.y, ps.L.z, lw);
    st.status[i] = (DEAD << 24) | (uint32_t)i;
}

// ----------------------------------------------------------------- splat
__global__ __launch_bounds__(256)
void k_wf_splat(SceneView sv, WfState st, float* __restrict__ accum, float* __restrict__ var, int nspp_done) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= st.n) return;
    float4 l4 = st.L[i];
    Vec3 L(l4.x, l4.y, l4.z);
    if (L.has_nan()) L = Vec3(0.f);
    L = clamp_radiance(sv, L);
    float* a = accum + (size_t)i * 4;
    a[0] += L.x; a[1] += L.y; a[2] += L.z; a[3] += 1.f;
    if (var) {
        float lum = (L.x + L.y + L.z) * (1.f / 3.f);
        var[(size_t)i * 2 + 0] += lum;
        var[(size_t)i * 2 + 1] += lum * lum;
    }
    (void)nspp_done;
}

// --------------------------------------------------------------- host side
template <typename T>
static int wf_alloc(T** p, size_t count) { return (int)hipMalloc((void**)p, count * sizeof(T)); }

WfState* wf_create(int width, int height) {
    WfState* s = new WfState();
    s->w = width; s->h = height; s->n = width * height;
    s->nb_sort = (s->n + SORT_BLOCK * SORT_ITEMS - 1) / (SORT_BLOCK * SORT_ITEMS);
    int e = 0;
    e |= wf_alloc(&s->ray_o, s->n);
    e |= wf_alloc(&s->ray_d, s->n);
    e |= wf_alloc(&s->thp, s->n);
    e |= wf_alloc(&s->L, s->n);
    e |= wf_alloc(&s->hit, s->n);
    e |= wf_alloc(&s->prevn, s->n);
    e |= wf_alloc(&s->rng, s->n);
    e |= wf_alloc(&s->status, s->n);
    e |= wf_alloc(&s->order, s->n);
    e |= wf_alloc(&s->order2, s->n);
    e |= wf_alloc(&s->hist, (size_t)s->nb_sort * 256);
    e |= wf_alloc(&s->live_dev, 1);
    e |= wf_alloc(&s->sh_od, s->n);
    e |= wf_alloc(&s->sh_dir, s->n);
    e |= wf_alloc(&s->sh_val, s->n);
    e |= wf_alloc(&s->sh_cnt, 1);
    e |= (int)hipHostMalloc((void**)&s->live_host, sizeof(int));
    if (e) { wf_destroy(s); return nullptr; }
    return s;
}

void wf_destroy(WfState* s) {
    if (!s) return;
    (void)hipFree(s->ray_o); (void)hipFree(s->ray_d); (void)hipFree(s->thp); (void)hipFree(s->L);
    (void)hipFree(s->hit); (void)hipFree(s->prevn); (void)hipFree(s->rng);
    (void)hipFree(s->status); (void)hipFree(s->order); (void)hipFree(s->order2);
    (void)hipFree(s->hist);
    (void)hipFree(s->live_dev);
    (void)hipFree(s->sh_od); (void)hipFree(s->sh_dir); (void)hipFree(s->sh_val); (void)hipFree(s->sh_cnt);
    if (s->live_host) (void)hipHostFree(s->live_host);
    delete s;
}

int launch_render_wavefront(WfState* st, const SceneView& sv, float* accum, float* var,
                            int spp0, int nspp, uint32_t seed, int sort_mode, void* stream) {
    if (sort_mode < 0 || sort_mode > 1) sort_mode = 0;
    static int env_mode = [] {
        const char* e = getenv("HIPPT_WF_SORT");
        return (e && strcmp(e, "compact") == 0) ? 1 : -1;
    }();
    if (env_mode >= 0) sort_mode = env_mode;
    hipStream_t hs = (hipStream_t)stream;
    const int n = st->n;
    dim3 blk(WF_BLOCK);
    dim3 grd_n((n + WF_BLOCK - 1) / WF_BLOCK);
    // HIPPT_WF_OCC = exact waves/SIMD for the traversal kernels (both the
    // amdgpu_waves_per_eu instantiation — frees 512/occ VGPRs so the walk
    // state does not spill — AND the matching LDS block budget).  Default 4,
    // the megakernel's measured best.  HIPPT_WF_STACK=scratch disables the
    // LDS stack + cache entirely (A/B hook).
    static int occ_v = [] {
        const char* e = getenv("HIPPT_WF_OCC");
        int occ = e ? atoi(e) : 4;   // span>=4 ladder: occ4 143-146 vs occ3 136
        return occ < 3 ? 3 : (occ > 6 ? 6 : occ);
    }();
    using RaygenFn = void (*)(SceneView, WfState, int, uint32_t, int, int);
    using PrimaryFn = void (*)(SceneView, WfState, int, uint32_t, int, int, int, int);
    using TraceFn = void (*)(SceneView, WfState, const uint32_t*, int, int, int);
    using ShadowFn = void (*)(SceneView, WfState, int, int);
    static RaygenFn f_raygen = occ_v == 3 ? k_wf_raygen<3> : occ_v == 4 ? k_wf_raygen<4>
                             : occ_v == 5 ? k_wf_raygen<5> : k_wf_raygen<6>;
    static PrimaryFn f_primary = occ_v == 3 ? k_wf_primary<3> : occ_v == 4 ? k_wf_primary<4>
                               : occ_v == 5 ? k_wf_primary<5> : k_wf_primary<6>;
    static int prim_fuse = [] {
        const char* e = getenv("HIPPT_WF_PRIMARY_FUSE");
        return e ? atoi(e) : 1;
    }();
    static TraceFn f_trace = occ_v == 3 ? k_wf_trace<3> : occ_v == 4 ? k_wf_trace<4>
                           : occ_v == 5 ? k_wf_trace<5> : k_wf_trace<6>;
    static TraceFn f_tail = occ_v == 3 ? k_wf_tail<3> : occ_v == 4 ? k_wf_tail<4>
                          : occ_v == 5 ? k_wf_tail<5> : k_wf_tail<6>;
    using StepFn = void (*)(SceneView, WfState, const uint32_t*, int, int, int, int, int);
    static StepFn f_step = occ_v == 3 ? k_wf_step<3> : occ_v == 4 ? k_wf_step<4>
                         : occ_v == 5 ? k_wf_step<5> : k_wf_step<6>;
    static int wf_fuse = [] {
        const char* e = getenv("HIPPT_WF_FUSE");
        return e ? atoi(e) : 1;
    }();
    static int wf_span = [] {
        // Shades per fused launch between sorts/compactions.  The kitchen
        // live-count curve (~99% live through bounce 6) makes long spans
        // pay: measured span1 124.5, span4 142.4, span8 145.9 Msps at occ4
        // — the optimal sort/compaction cadence on this latency-bound
        // hardware is about once per 8 bounces.  span=1 restores the
        // classic per-bounce wavefront.
        const char* e = getenv("HIPPT_WF_SPAN");
        int v = e ? atoi(e) : 8;
        return v < 1 ? 1 : (v > 16 ? 16 : v);
    }();
    static ShadowFn f_shadow = occ_v == 3 ? k_wf_shadow<3> : occ_v == 4 ? k_wf_shadow<4>
                             : occ_v == 5 ? k_wf_shadow<5> : k_wf_shadow<6>;
    static int lds_budget = [] {
        const char* e = getenv("HIPPT_WF_STACK");
        if (e && strcmp(e, "scratch") == 0) return 0;
        return (occ_v == 3 ? 26 : occ_v == 4 ? 20 : occ_v == 5 ? 16 : 12) * WF_BLOCK * 8;
    }();
    // LDS top-tree cache shares the block budget with the stacks (same
    // occupancy, fewer LDS stack entries; overflow spills to scratch)
    static int cache_env = [] {
        const char* e = getenv("HIPPT_TOPCACHE");
        return e ? atoi(e) : -1;
    }();
    int cache_req = cache_env >= 0 ? cache_env
                  : (sv.cache_nodes > 0 ? sv.cache_nodes : 64);
    if (cache_req * (int)sizeof(TravNode) > lds_budget - 4 * WF_BLOCK * 8)
        cache_req = (lds_budget - 4 * WF_BLOCK * 8) / (int)sizeof(TravNode);
    const int n_cached = lds_budget > 0
        ? (cache_req < sv.n_nodes4 ? cache_req : sv.n_nodes4) : 0;
    const int lds_n = lds_budget > 0
        ? (lds_budget - n_cached * (int)sizeof(TravNode)) / (WF_BLOCK * 8) : 0;
    const uint32_t shmem = (uint32_t)(lds_n * WF_BLOCK * 8 + n_cached * (int)sizeof(TravNode));
    static int wf_dual = [] {
        const char* e = getenv("HIPPT_WF_DUAL");
        return e ? atoi(e) : 0;
    }();
    // dual-walk stack safety (each walk: BVH4_STACK/2 private + lds_n/2 LDS
    // entries; a node visit pushes at most 3 entries per level)
    if (wf_dual && 3 * sv.bvh4_depth > BVH4_STACK / 2 + lds_n / 2) wf_dual = 0;
    static int tail_thresh = [] {
        // Hand off to the fused tail once <=1.5M rays are live (measured
        // best on kitchen 1080p: per-bounce pipeline overhead outweighs the
        // wave64 divergence it avoids well before the chip runs dry).
        const char* e = getenv("HIPPT_WF_TAIL");
        return e ? atoi(e) : 1500 * 1024;
    }();
    static int wf_log = [] {
        const char* e = getenv("HIPPT_WF_LOG");
        return e ? atoi(e) : 0;
    }();
    for (int s = 0; s < nspp; ++s) {
        int bounce0;
        const int n_stage = sv.md.max_depth + 1;   // total shades per path
        if (prim_fuse && wf_fuse) {
            // camera ray + primary hit + span shades + next trace fused
            int cnt = wf_span < n_stage ? wf_span : n_stage;
            hipLaunchKernelGGL(f_primary, grd_n, blk, shmem, hs, sv, *st, spp0 + s,
                               seed, cnt - 1, cnt < n_stage ? 1 : 0, lds_n, n_cached);
            bounce0 = cnt;
        } else {
            hipLaunchKernelGGL(f_raygen, grd_n, blk, shmem, hs, sv, *st, spp0 + s, seed,
                               lds_n, n_cached);
            bounce0 = 0;
        }
        // Per bounce: build the compacted material-sorted live view, read the
        // live count back (4 bytes — the price the reference also pays,
        // wf_path_tracer.cu:199), then launch live-sized grids.  Bounce 0
        // sorts the full pixel-order status array; later bounces gather
        // through the previous view, so the sort scales with live too.
        const uint32_t* gather = nullptr;
        int prev_live = n;
        uint32_t* order_cur = st->order;
        uint32_t* order_prev = st->order2;
        for (int bounce = bounce0; bounce < n_stage; bounce += (wf_fuse ? wf_span : 1)) {
            const int scan_n = prev_live;
            const int nb = (scan_n + SORT_BLOCK * SORT_ITEMS - 1) / (SORT_BLOCK * SORT_ITEMS);
            (void)hipMemsetAsync(st->hist, 0, 256 * sizeof(uint32_t), hs);
            hipLaunchKernelGGL(k_sort_hist, dim3(nb), dim3(SORT_BLOCK), 0, hs,
                               st->status, gather, scan_n, st->hist, sort_mode);
            hipLaunchKernelGGL(k_sort_scan, dim3(1), dim3(256), 0, hs,
                               st->hist, st->live_dev);
            hipLaunchKernelGGL(k_sort_scatter, dim3(nb), dim3(SORT_BLOCK), 0, hs,
                               st->status, gather, scan_n, st->hist, order_cur, sort_mode);
            (void)hipMemcpyAsync(st->live_host, st->live_dev, sizeof(int),
                                 hipMemcpyDeviceToHost, hs);
            int err = (int)hipStreamSynchronize(hs);
            if (err) return err;
            const int live = *st->live_host;
            if (wf_log && s == 0)
                printf("[wf] spp %d bounce %d live %d (%.1f%%)\n", spp0 + s, bounce,
                       live, 100.0 * live / n);
            if (live == 0) break;                     // reference: break at live==0
            dim3 grd_live((live + WF_BLOCK - 1) / WF_BLOCK);
            if (live <= tail_thresh) {
                hipLaunchKernelGGL(f_tail, grd_live, blk, shmem, hs, sv, *st,
                                   order_cur, live, lds_n, n_cached);
                break;
            }
            if (wf_fuse) {
                int cnt = wf_span < n_stage - bounce ? wf_span : n_stage - bounce;
                hipLaunchKernelGGL(f_step, grd_live, blk, shmem, hs, sv, *st, order_cur,
                                   live, cnt - 1, bounce + cnt < n_stage ? 1 : 0,
                                   lds_n, n_cached);
            } else {
            (void)hipMemsetAsync(st->sh_cnt, 0, sizeof(int), hs);
            hipLaunchKernelGGL(k_wf_shade, grd_live, blk, 0, hs, sv, *st, order_cur,
                               live, bounce);
            hipLaunchKernelGGL(f_shadow, grd_live, blk, shmem, hs, sv, *st,
                               lds_n, n_cached);
            if (bounce < sv.md.max_depth) {           // reference skips last-bounce trace
                if (wf_dual) {
                    dim3 grd_h(((live + 1) / 2 + WF_BLOCK - 1) / WF_BLOCK);
                    hipLaunchKernelGGL(k_wf_trace_dual, grd_h, blk,
                                       (uint32_t)(lds_n * WF_BLOCK * 8), hs, sv, *st,
                                       order_cur, live, lds_n);
                } else {
                    hipLaunchKernelGGL(f_trace, grd_live, blk, shmem, hs, sv, *st,
                                       order_cur, live, lds_n, n_cached);
                }
            }
            }
            gather = order_cur;
            prev_live = live;
            uint32_t* t = order_cur; order_cur = order_prev; order_prev = t;
        }
        hipLaunchKernelGGL(k_wf_splat, grd_n, blk, 0, hs, sv, *st, accum, var, 1);
    }
    return (int)hipGetLastError();
}

} // namespace hippt
