// cpu_render.cpp — the CPU reference renderer: same integrator source as the
// GPU kernels, parallelized over pixel rows with std::thread.
//
// This is the "pyrender CPU host path" of BASELINE config #1 and the ground
// truth that GPU numerics tests compare against (tests/test_gpu_*.py).
#include "../core/integrator.h"
#include "../core/light_tracer.h"
#include "../core/integrator_vol.h"
#include <thread>
#include <functional>
#include <atomic>
#include <vector>

namespace hippt {

static void parallel_rows(int h, int n_threads, const std::function<void(int)>& fn) {
    std::atomic<int> next{0};
    int nt = std::max(1u, std::min((unsigned)n_threads, std::thread::hardware_concurrency()));
    std::vector<std::thread> ts;
    for (int t = 0; t < nt; ++t)
        ts.emplace_back([&] {
            for (;;) {
                int y = next.fetch_add(1);
                if (y >= h) return;
                fn(y);
            }
        });
    for (auto& t : ts) t.join();
}

// Accumulate `nspp` samples into accum (h,w,4 float32: RGB sum + sample count)
// and var (h,w,2 float32: lum sum, lum^2 sum).  renderer: 0=PT, 2=VPT,
// 4=depth, 5=bvh_cost (light tracing has its own entry below).
void render_cpu(const SceneView& sv, float* accum, float* var,
                int spp0, int nspp, uint32_t seed, int renderer, int n_threads,
                int y0, int y1, const uint8_t* spp_map, float* aux) {
    const int w = sv.cam.w, h = sv.cam.h;
    if (y1 <= 0 || y1 > h) y1 = h;
    if (y0 < 0) y0 = 0;
    parallel_rows(y1 - y0, n_threads, [&](int yr) {
        int y = yr + y0;
        for (int x = 0; x < w; ++x) {
            size_t pix = size_t(y) * w + x;
            const int nspp_px = spp_map ? (int)spp_map[pix] : nspp;
            if (nspp_px == 0) continue;
            Vec3 Lsum(0.f);
            float lum_s = 0.f, lum_s2 = 0.f;
            Vec3 an(0.f), aa(0.f);
            float at = 0.f;
            for (int s = 0; s < nspp_px; ++s) {
                Sampler sp(uint32_t(pix), uint32_t(spp0 + s) * SEED_SCALER + seed);
                Ray ray = sv.cam.gen_ray(x, y, sp, spp0 + s);
                Vec3 L(0.f);
                if (renderer == 2)      L = clamp_radiance(sv, trace_path_volumetric(sv, ray, sp));
                else if (renderer == 4) L = Vec3(trace_depth(sv, ray));
                else if (renderer == 5) { Vec2 c = trace_bvh_cost(sv, ray); L = Vec3(c.x, c.y, 0.f); }
                else if (aux) {
                    PathState ps;
                    ps.reset(ray);
                    while (!path_step(sv, ps, sp, TravCtx{})) {}
                    L = ps.L.has_nan() ? Vec3(0.f) : clamp_radiance(sv, ps.L);
                    an += ps.aov_n; aa += ps.aov_alb; at += ps.aov_t;
                } else                  L = trace_path(sv, ray, sp);
                Lsum += L;
                float lum = (L.x + L.y + L.z) * (1.f / 3.f);
                lum_s += lum; lum_s2 += lum * lum;
            }
            if (aux) {
                float* a8 = aux + pix * 8;
                a8[0] += an.x; a8[1] += an.y; a8[2] += an.z; a8[3] += at;
                a8[4] += aa.x; a8[5] += aa.y; a8[6] += aa.z; a8[7] += (float)nspp_px;
            }
            accum[pix * 4 + 0] += Lsum.x;
            accum[pix * 4 + 1] += Lsum.y;
            accum[pix * 4 + 2] += Lsum.z;
            accum[pix * 4 + 3] += (float)nspp_px;
            if (var) { var[pix * 2 + 0] += lum_s; var[pix * 2 + 1] += lum_s2; }
        }
    });
}

// Light tracing pass: traces nspp light paths per pixel-equivalent budget and
// splats to the image with atomics-free per-thread accumulation + merge.
void render_lt_cpu(const SceneView& sv, float* accum, int spp0, int nspp, uint32_t seed,
                   int spec_constraint, float caustic_scaling, int n_threads) {
    const int w = sv.cam.w, h = sv.cam.h;
    const size_t npix = size_t(w) * h;
    int nt = std::max(1u, std::min((unsigned)n_threads, std::thread::hardware_concurrency()));
    std::vector<std::vector<float>> partial(nt, std::vector<float>(npix * 4, 0.f));
    std::vector<std::thread> ts;
    std::atomic<int> chunk{0};
    const int n_chunks = nt * 8;
    const long long total_paths = (long long)npix * nspp;
    for (int t = 0; t < nt; ++t)
        ts.emplace_back([&, t] {
            float* img = partial[t].data();
            for (;;) {
                int c = chunk.fetch_add(1);
                if (c >= n_chunks) return;
                long long lo = total_paths * c / n_chunks, hi = total_paths * (c + 1) / n_chunks;
                for (long long i = lo; i < hi; ++i) {
                    Sampler sp(uint32_t(i & 0xffffffff), uint32_t(spp0) * SEED_SCALER + seed + uint32_t(i >> 32));
                    trace_light_path(sv, sp, img, w, h, spec_constraint, caustic_scaling);
                }
            }
        });
    for (auto& th : ts) th.join();
    for (int t = 0; t < nt; ++t)
        for (size_t i = 0; i < npix * 4; ++i) accum[i] += partial[t][i];
    // per-pixel path count normalization: each pixel's share of paths
    for (size_t i = 0; i < npix; ++i) accum[i * 4 + 3] += (float)nspp;
}

} // namespace hippt
