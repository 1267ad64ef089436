// kernels.h — launch API of the gfx950 HIP kernels (implemented in *.hip).
#pragma once
#include <cstdint>
#include "../core/scene_view.h"

namespace hippt {

enum RendererKind : int {
    R_MEGAKERNEL_PT = 0,
    R_WAVEFRONT_PT = 1,
    R_VOLUME_PT = 2,
    R_LIGHT_TRACE = 3,
    R_DEPTH = 4,
    R_BVH_COST = 5,
    R_MEGAKERNEL_PT_DYN = 6,
};

// Accumulate nspp samples into accum (h*w*4: RGB sum + count) and var
// (h*w*2: lum sum, lum^2 sum). sv must hold DEVICE pointers.
// Returns hipError_t as int.
// y0/y1: optional row band [y0, y1) for tile-split DP (0,0 = full frame);
// applies to the megakernel renderers (wavefront/lt always render full).
// spp_map: optional per-pixel sample budget for this launch (adaptive
// sampling; overrides nspp per pixel, 0 = skip pixel).
int launch_render(const SceneView& sv, float* accum, float* var,
                  int spp0, int nspp, uint32_t seed, int renderer,
                  int spec_constraint, float caustic_scaling,
                  void* stream, int y0 = 0, int y1 = 0,
                  const uint8_t* spp_map = nullptr,
                  float* aux = nullptr);  // (h*w*8) primary-hit AOV sums:
                                          // n.xyz, depth, albedo.rgb, count

// Wavefront path tracer (SoA queues + compaction); state owned by WfState.
struct WfState;
WfState* wf_create(int width, int height);
void wf_destroy(WfState* s);
int launch_render_wavefront(WfState* st, const SceneView& sv, float* accum, float* var,
                            int spp0, int nspp, uint32_t seed, int sort_mode, void* stream);

// device helpers used by bind.cpp
int dev_malloc(void** p, size_t n);
int dev_free(void* p);
int dev_upload(void* dst, const void* src, size_t n);
int dev_memset(void* dst, int v, size_t n, void* stream);
int dev_synchronize();
int dev_set_device(int d);

} // namespace hippt
