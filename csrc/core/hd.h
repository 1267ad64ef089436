// hd.h — single-source host/device qualifiers for the hippt core.
//
// The whole rendering core (math, BSDFs, BVH traversal, integrators) is
// written once in these headers and compiled twice:
//   * as plain C++ (OpenMP) for the CPU reference renderer / tests, and
//   * as HIP device code for the gfx950 (MI355X, wave64) kernels.
// No CUDA compatibility layer; HIP-only device path.
//
// Capability parity target: reference src/core/cuda_utils.cuh (CPT_* macros).
#pragma once

#include <cstdint>
#include <cmath>
#include <cstring>

#if defined(__HIPCC__)
#include <hip/hip_runtime.h>
#define HD __host__ __device__ __forceinline__
#define HOSTFN __host__ inline
#else
#define HD inline
#define HOSTFN inline
#endif

#if defined(__HIP_DEVICE_COMPILE__)
#define HIPPT_ON_DEVICE 1
#else
#define HIPPT_ON_DEVICE 0
#endif

namespace hippt {

constexpr float EPSILON      = 1e-3f;   // shadow/offset epsilon (reference src/core/constants.cuh)
constexpr float THP_EPS      = 1e-5f;
constexpr float MAX_DIST     = 1e7f;
constexpr float ENVMAP_DIST  = 5e3f;
constexpr float PI           = 3.14159265358979323846f;
constexpr float INV_PI       = 0.31830988618379067154f;
constexpr float TWO_PI       = 6.28318530717958647692f;
constexpr uint32_t WAVE_SIZE = 64;      // CDNA4 wavefront

HD float f_as_u(float f, uint32_t* u) { // no-op helper kept for clarity
    std::memcpy(u, &f, 4); return f;
}
HD uint32_t float_as_uint(float f) { uint32_t u; std::memcpy(&u, &f, 4); return u; }
HD float uint_as_float(uint32_t u) { float f; std::memcpy(&f, &u, 4); return f; }
HD int32_t float_as_int(float f) { int32_t i; std::memcpy(&i, &f, 4); return i; }
HD float int_as_float(int32_t i) { float f; std::memcpy(&f, &i, 4); return f; }

template <typename T> HD T clampv(T v, T lo, T hi) { return v < lo ? lo : (v > hi ? hi : v); }
HD float saturatef(float v) { return clampv(v, 0.f, 1.f); }

// branchless select (reference cuda_utils.cuh `select`)
template <typename T> HD T select(bool c, T a, T b) { return c ? a : b; }

} // namespace hippt
