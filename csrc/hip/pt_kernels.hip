// pt_kernels.hip — gfx950 (MI355X, wave64) megakernel renderers.
//
// Capability parity: reference src/pt_impl/megakernel_pt.cu (render_pt_kernel),
// megakernel_vpt.cu (render_vpt_kernel), megakernel_lt.cu (render_lt_kernel),
// depth.cu, bvh_cost.cu.  Kernels are written directly against CDNA4:
// 256-thread blocks (4 wave64), 16x16 pixel tiles so each wave covers a
// coherent 16x4 strip, grids of thousands of workgroups to fill 256 CUs
// across 8 XCDs, and a per-kernel spp loop to amortize launch overhead.
#include <hip/hip_runtime.h>
#include "kernels.h"
#include <cstdlib>
#include <cstring>
#include "../core/integrator.h"
#include "../core/integrator_vol.h"
#include "../core/light_tracer.h"

namespace hippt {

// ------------------------------------------------------------- PT megakernel
// MINW = exact waves/SIMD residency (amdgpu_waves_per_eu pin; HIPPT_OCC
// selects the instantiation at launch).
// XCD-aware tile swizzle: MI355X dispatches consecutive workgroups
// round-robin over the 8 XCDs (each with its own L2).  Remapping the flat
// block id so blocks congruent mod 8 cover one contiguous screen band gives
// each XCD a compact working set (one band's BVH leaves + prims) instead of
// a scatter of the whole frame.  Bijection on [0, n_tiles):
//   xcd = flat % 8, id = xcd*(n/8) + min(xcd, n%8) + flat/8.
__device__ inline int xcd_swizzle(int flat, int n_tiles) {
    int xcd = flat & 7, local = flat >> 3;
    int per = n_tiles >> 3, rem = n_tiles & 7;
    return xcd * per + (xcd < rem ? xcd : rem) + local;
}

// lds_n = traversal-stack entries per thread held in (dynamic) LDS, bvh4.h.
// The LDS block allocation (lds_n x 8 B x 256 threads) is also the occupancy
// governor: 12/16/20/26/40 entries -> 6/5/4/3/2 waves per SIMD.
template <int RENDERER, int MINW = 6>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))   // exact residency: the LDS
// stack caps blocks/CU at MINW waves/SIMD, so the allocator may spend the
// full 512/MINW VGPRs instead of hoarding occupancy it can't get (measured:
// plain launch_bounds left the kernel at 40 VGPR + ~1 KB/lane of spills)
void k_render(SceneView sv, float* __restrict__ accum, float* __restrict__ var,
              int spp0, int nspp, uint32_t seed, int swiz, int lds_n, int n_cached, int y_off, int y_end,
              const uint8_t* __restrict__ spp_map, float* __restrict__ aux) {
    extern __shared__ uint64_t s_stk[];
    // dynamic-LDS layout: [n_cached 128-byte nodes][per-thread stacks].
    // The top of the tree is copied into LDS once per block: every walk's
    // first visits are nodes 0..n_cached, and at a 95% L2 hit rate the
    // bound is hit latency — LDS is ~4x closer.
    constexpr int NW = (int)(sizeof(TravNode) / 8);   // u64 words per node
    TravNode* s_cache = (TravNode*)s_stk;
    uint64_t* s_base = s_stk + (size_t)n_cached * NW;
    const int tid = threadIdx.y * 16 + threadIdx.x;
    for (int i = tid; i < n_cached * NW; i += 256)
        ((uint64_t*)s_cache)[i] = ((const uint64_t*)trav_nodes(sv))[i];
    if (n_cached > 0) __syncthreads();
    TravCtx tc{&s_base[tid], lds_n, s_cache, n_cached};
    int tile = blockIdx.y * gridDim.x + blockIdx.x;
    if (swiz) tile = xcd_swizzle(tile, gridDim.x * gridDim.y);
    const int px = (tile % gridDim.x) * 16 + threadIdx.x;
    const int py = (tile / gridDim.x) * 16 + threadIdx.y + y_off;
    if (px >= sv.cam.w || py >= y_end) return;   // y_end = band end (<= h)
    const size_t pix = size_t(py) * sv.cam.w + px;

    // adaptive sampling: per-pixel sample budget for this launch (variance
    // routed back into work allocation; accumulator alpha carries per-pixel
    // counts, so heterogeneous spp still averages exactly)
    const int nspp_px = spp_map ? (int)spp_map[pix] : nspp;
    if (nspp_px == 0) return;
    Vec3 Lsum(0.f);
    float lum_s = 0.f, lum_s2 = 0.f;
    if constexpr (RENDERER == R_MEGAKERNEL_PT) {
        // Path regeneration: the lane starts its next sample the moment its
        // path dies, instead of idling until the wave's longest path ends
        // (wave64 tail divergence was the measured bottleneck).  Sampler
        // streams and results are identical to the per-sample loop.
        int s = 0;
        Vec3 an(0.f), aa(0.f);
        float at = 0.f;
        Sampler sp(uint32_t(pix), uint32_t(spp0) * SEED_SCALER + seed);
        PathState ps;
        ps.reset(sv.cam.gen_ray(px, py, sp, spp0));
        for (;;) {
            if (path_step(sv, ps, sp, tc)) {
                Vec3 L = ps.L.has_nan() ? Vec3(0.f) : clamp_radiance(sv, ps.L);
                Lsum += L;
                float lum = (L.x + L.y + L.z) * (1.f / 3.f);
                lum_s += lum;
                lum_s2 = fmaf(lum, lum, lum_s2);
                if (aux) { an += ps.aov_n; aa += ps.aov_alb; at += ps.aov_t; }
                if (++s >= nspp_px) break;
                sp = Sampler(uint32_t(pix), uint32_t(spp0 + s) * SEED_SCALER + seed);
                ps.reset(sv.cam.gen_ray(px, py, sp, spp0 + s));
            }
        }
        if (aux) {
            float* a8 = aux + pix * 8;
            a8[0] += an.x; a8[1] += an.y; a8[2] += an.z; a8[3] += at;
            a8[4] += aa.x; a8[5] += aa.y; a8[6] += aa.z; a8[7] += (float)nspp_px;
        }
    } else
    for (int s = 0; s < nspp_px; ++s) {
        Sampler sp(uint32_t(pix), uint32_t(spp0 + s) * SEED_SCALER + seed);
        Ray ray = sv.cam.gen_ray(px, py, sp, spp0 + s);
        Vec3 L(0.f);
        if constexpr (RENDERER == R_VOLUME_PT) L = clamp_radiance(sv, trace_path_volumetric(sv, ray, sp, tc));
        else if constexpr (RENDERER == R_DEPTH) L = Vec3(trace_depth(sv, ray, tc));
        else if constexpr (RENDERER == R_BVH_COST) { Vec2 c = trace_bvh_cost(sv, ray); L = Vec3(c.x, c.y, 0.f); }
        else L = trace_path(sv, ray, sp, tc);
        Lsum += L;
        float lum = (L.x + L.y + L.z) * (1.f / 3.f);
        lum_s += lum;
        lum_s2 = fmaf(lum, lum, lum_s2);
    }
    float* a = accum + pix * 4;
    a[0] += Lsum.x; a[1] += Lsum.y; a[2] += Lsum.z; a[3] += (float)nspp_px;
    if (var) { var[pix * 2 + 0] += lum_s; var[pix * 2 + 1] += lum_s2; }
}

// ------------------------------------------------- persistent-tile variant
// Capability parity: reference scheduler.cuh:75-96 PreemptivePersistentTileScheduler
// (persistent blocks atomically grab the next tile id from a global counter).
// On MI355X: grid = 256 CUs x blocks/CU, tile = 16x16 pixels, counter in
// device memory zeroed per launch; removes the tail effect of uneven
// per-tile path lengths on the 8-XCD chip.
template <int RENDERER, int MINW = 6>
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(MINW, MINW)))
void k_render_persistent(SceneView sv, float* __restrict__ accum, float* __restrict__ var,
                         int spp0, int nspp, uint32_t seed, uint32_t* work_counter,
                         int tiles_x, int n_tiles, int lds_n) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc{&s_stk[threadIdx.y * 16 + threadIdx.x], lds_n};
    __shared__ uint32_t s_tile;
    for (;;) {
        if (threadIdx.x == 0 && threadIdx.y == 0)
            s_tile = atomicAdd(work_counter, 1u);
        __syncthreads();
        uint32_t tile = s_tile;
        __syncthreads();
        if (tile >= (uint32_t)n_tiles) return;
        int tx = (int)(tile % tiles_x), ty = (int)(tile / tiles_x);
        int px = tx * 16 + threadIdx.x;
        int py = ty * 16 + threadIdx.y;
        if (px >= sv.cam.w || py >= sv.cam.h) continue;
        const size_t pix = size_t(py) * sv.cam.w + px;
        Vec3 Lsum(0.f);
        float lum_s = 0.f, lum_s2 = 0.f;
        // per-sample loop (path regen measured -14% here: persistent blocks
        // already smooth tile tails, and regen lengthens the per-tile stint)
        for (int s = 0; s < nspp; ++s) {
            Sampler sp(uint32_t(pix), uint32_t(spp0 + s) * SEED_SCALER + seed);
            Ray ray = sv.cam.gen_ray(px, py, sp, spp0 + s);
            Vec3 L = trace_path(sv, ray, sp, tc);
            Lsum += L;
            float lum = (L.x + L.y + L.z) * (1.f / 3.f);
            lum_s += lum;
            lum_s2 = fmaf(lum, lum, lum_s2);
        }
        float* a = accum + pix * 4;
        a[0] += Lsum.x; a[1] += Lsum.y; a[2] += Lsum.z; a[3] += (float)nspp;
        if (var) { var[pix * 2 + 0] += lum_s; var[pix * 2 + 1] += lum_s2; }
    }
}

// ------------------------------------------------------------- light tracing
__global__ __launch_bounds__(256)
__attribute__((amdgpu_waves_per_eu(6, 6)))
void k_render_lt(SceneView sv, float* __restrict__ accum,
                 long long n_paths, int spp0, int nspp, uint32_t seed,
                 int spec_constraint, float caustic_scaling, int lds_n) {
    extern __shared__ uint64_t s_stk[];
    TravCtx tc{&s_stk[threadIdx.x], lds_n};
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n_paths; i += stride) {
        Sampler sp(uint32_t(i & 0xffffffff), uint32_t(spp0) * SEED_SCALER + seed + uint32_t(i >> 32));
        trace_light_path_impl(sv, sp, [&](int pix, Vec3 v) {
            atomicAdd(accum + pix * 4 + 0, v.x);
            atomicAdd(accum + pix * 4 + 1, v.y);
            atomicAdd(accum + pix * 4 + 2, v.z);
        }, tc, spec_constraint, caustic_scaling);
    }
}

__global__ void k_add_count(float* __restrict__ accum, size_t npix, float cnt) {
    size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < npix) accum[i * 4 + 3] += cnt;
}

// ------------------------------------------------------------------ launches
int launch_render(const SceneView& sv, float* accum, float* var,
                  int spp0, int nspp, uint32_t seed, int renderer,
                  int spec_constraint, float caustic_scaling, void* stream,
                  int y0, int y1, const uint8_t* spp_map, float* aux) {
    hipStream_t st = (hipStream_t)stream;
    // HIPPT_SWIZZLE=1 enables the XCD band swizzle (measured -14% on the
    // kitchen megakernel — the round-robin XCD dispatch already spreads
    // neighbor tiles well; default off, kept as an A/B hook)
    static int swiz_v = [] {
        const char* e = getenv("HIPPT_SWIZZLE");
        return e ? atoi(e) : 0;
    }();
    auto swiz = [] { return swiz_v; };
    // Two decoupled knobs (defaults from same-box A/B on MI355X, see
    // profiles/README.md):
    //   HIPPT_OCC   = __launch_bounds__ min waves/SIMD {3,4,5,6,8}.  The
    //   register CAP is what the walk wants (waves 6 spills VGPRs yet was
    //   measured ~1.5x over the compiler's natural allocation).  Default 6.
    //   HIPPT_STACK = lds | scratch: where the BVH4 traversal stack lives
    //   (LDS entries sized {26,20,16,12,12} for occ {3,4,5,6,8}).
    static int occ_v = [] {
        const char* e = getenv("HIPPT_OCC");
        return e ? atoi(e) : 4;   // ww walk: occ4 measured 140.6 vs 128.9 @6
    }();
    // HIPPT_TOPCACHE = nodes of the tree top copied to LDS per block
    // (default 64 = 8 KB); the stack's LDS share shrinks to keep the same
    // per-block LDS budget (occupancy unchanged).
    static int cache_env = [] {
        const char* e = getenv("HIPPT_TOPCACHE");
        return e ? atoi(e) : -1;
    }();
    // priority: env A/B hook > accelerator XML cache_level > measured default
    int cache_req = cache_env >= 0 ? cache_env
                  : (sv.cache_nodes > 0 ? sv.cache_nodes : 64);
    int n_cached = cache_req < sv.n_nodes4 ? cache_req : sv.n_nodes4;
    // keep at least 4 LDS stack entries per thread alongside the cache
    int cache_cap = (20 * 256 * 8 - 4 * 256 * 8) / (int)sizeof(TravNode);
    if (n_cached > cache_cap) n_cached = cache_cap;
    static int lds_budget = [] {
        const char* e = getenv("HIPPT_STACK");
        if (e && strcmp(e, "scratch") == 0) return 0;   // default: lds
        return (occ_v <= 3 ? 26 : occ_v == 4 ? 20 : occ_v == 5 ? 16 : 12) * 256 * 8;
    }();
    const int lds_n = lds_budget > 0
        ? (lds_budget - n_cached * (int)sizeof(TravNode)) / (256 * 8) : 0;
    const uint32_t shmem = (uint32_t)(lds_n * 256 * 8 + n_cached * (int)sizeof(TravNode));
    const int w = sv.cam.w, h = sv.cam.h;
    if (y1 <= 0 || y1 > h) y1 = h;
    if (y0 < 0) y0 = 0;
    dim3 block(16, 16);
    dim3 grid((w + 15) / 16, (y1 - y0 + 15) / 16);
    switch (renderer) {
    case R_LIGHT_TRACE: {
        long long n_paths = (long long)w * h * nspp;
        int nblk = 256 * 8 * 4;  // 256 CUs x enough blocks to fill + stride
        hipLaunchKernelGGL(k_render_lt, dim3(nblk), dim3(256), shmem, st,
                           sv, accum, n_paths, spp0, nspp, seed, spec_constraint, caustic_scaling, lds_n);
        size_t npix = (size_t)w * h;
        hipLaunchKernelGGL(k_add_count, dim3((npix + 255) / 256), dim3(256), 0, st,
                           accum, npix, (float)nspp);
        break;
    }
    case R_VOLUME_PT:
        if (occ_v == 4)
            hipLaunchKernelGGL((k_render<R_VOLUME_PT, 4>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        else
            hipLaunchKernelGGL((k_render<R_VOLUME_PT, 6>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        break;
    case R_DEPTH:
        hipLaunchKernelGGL((k_render<R_DEPTH, 6>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, 0, lds_n, n_cached, y0, y1, spp_map, aux);
        break;
    case R_BVH_COST:
        hipLaunchKernelGGL((k_render<R_BVH_COST, 6>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, 0, lds_n, n_cached, y0, y1, spp_map, aux);
        break;
    case R_MEGAKERNEL_PT_DYN: {
        // per-device work counters, freed at process exit (one process may
        // drive several devices: one slot per device id)
        static struct Counters {
            uint32_t* p[64] = {};
            ~Counters() { for (uint32_t* q : p) if (q) (void)hipFree(q); }
        } counters;
        int dev = 0;
        (void)hipGetDevice(&dev);
        uint32_t*& counter = counters.p[dev & 63];
        if (!counter) {
            if (hipMalloc((void**)&counter, 4) != hipSuccess) return (int)hipGetLastError();
        }
        (void)hipMemsetAsync(counter, 0, 4, st);
        int tiles_x = (w + 15) / 16, tiles_y = (h + 15) / 16;
        // 256 CUs x 4 blocks/CU (lds_n=20 caps residency at 4 blocks/CU)
        hipLaunchKernelGGL((k_render_persistent<R_MEGAKERNEL_PT, 6>), dim3(256 * 4), block, shmem, st,
                           sv, accum, var, spp0, nspp, seed, counter, tiles_x,
                           tiles_x * tiles_y, lds_n);
        break;
    }
    default: {
        if (occ_v <= 3)
            hipLaunchKernelGGL((k_render<R_MEGAKERNEL_PT, 3>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        else if (occ_v == 4)
            hipLaunchKernelGGL((k_render<R_MEGAKERNEL_PT, 4>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        else if (occ_v == 5)
            hipLaunchKernelGGL((k_render<R_MEGAKERNEL_PT, 5>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        else if (occ_v >= 8)
            hipLaunchKernelGGL((k_render<R_MEGAKERNEL_PT, 8>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        else
            hipLaunchKernelGGL((k_render<R_MEGAKERNEL_PT, 6>), grid, block, shmem, st, sv, accum, var, spp0, nspp, seed, swiz(), lds_n, n_cached, y0, y1, spp_map, aux);
        break;
    }
    }
    return (int)hipGetLastError();
}

// ------------------------------------------------------------- device utils
int dev_malloc(void** p, size_t n) { return (int)hipMalloc(p, n); }
int dev_free(void* p) { return (int)hipFree(p); }
int dev_upload(void* dst, const void* src, size_t n) {
    return (int)hipMemcpy(dst, src, n, hipMemcpyHostToDevice);
}
int dev_memset(void* dst, int v, size_t n, void* stream) {
    return (int)hipMemsetAsync(dst, v, n, (hipStream_t)stream);
}
int dev_synchronize() { return (int)hipDeviceSynchronize(); }
int dev_set_device(int d) { return (int)hipSetDevice(d); }

} // namespace hippt
