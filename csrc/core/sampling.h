// sampling.h — primitive samplers (cosine hemisphere, sphere, cone, disk,
// triangle) with pdf out-params.
//
// Capability parity: reference src/core/sampling.cuh:27-67.
#pragma once
#include "frame.h"

namespace hippt {

// Cosine-weighted hemisphere around +z. pdf = cos/pi.
HD Vec3 sample_cosine_hemisphere(Vec2 u, float& pdf) {
    float r = sqrtf(u.x);
    float phi = TWO_PI * u.y;
    float x = r * cosf(phi), y = r * sinf(phi);
    float z = sqrtf(fmaxf(0.f, 1.f - u.x));
    pdf = z * INV_PI;
    return {x, y, z};
}

// Uniform unit sphere. pdf = 1/(4 pi).
HD Vec3 sample_uniform_sphere(Vec2 u, float& pdf) {
    float z = 1.f - 2.f * u.x;
    float r = sqrtf(fmaxf(0.f, 1.f - z * z));
    float phi = TWO_PI * u.y;
    pdf = 1.f / (4.f * PI);
    return {r * cosf(phi), r * sinf(phi), z};
}

// Uniform cone around +z with cos(half-angle)=cos_max. pdf = 1/(2pi(1-cos_max)).
HD Vec3 sample_uniform_cone(Vec2 u, float cos_max, float& pdf) {
    float cos_t = (1.f - u.x) + u.x * cos_max;
    float sin_t = sqrtf(fmaxf(0.f, 1.f - cos_t * cos_t));
    float phi = TWO_PI * u.y;
    pdf = 1.f / (TWO_PI * fmaxf(1e-8f, 1.f - cos_max));
    return {sin_t * cosf(phi), sin_t * sinf(phi), cos_t};
}

// Concentric disk sample (for thin-lens DoF).
HD Vec2 sample_concentric_disk(Vec2 u) {
    float ox = 2.f * u.x - 1.f, oy = 2.f * u.y - 1.f;
    if (ox == 0.f && oy == 0.f) return {0.f, 0.f};
    float r, theta;
    if (fabsf(ox) > fabsf(oy)) { r = ox; theta = (PI / 4.f) * (oy / ox); }
    else                       { r = oy; theta = (PI / 2.f) - (PI / 4.f) * (ox / oy); }
    return {r * cosf(theta), r * sinf(theta)};
}

// Uniform barycentric point in a triangle (sqrt warp).
HD Vec3 sample_triangle_bary(Vec2 u) {
    float su = sqrtf(u.x);
    float b1 = 1.f - su, b2 = u.y * su;
    return {1.f - b1 - b2, b1, b2};
}

// MIS balance-heuristic weight for strategy a vs b (power=1, matching the
// reference's emit_len_mis semantics in megakernel_pt.cu:141-151).
HD float mis_weight(float pdf_a, float pdf_b) {
    float s = pdf_a + pdf_b;
    return s > 0.f ? pdf_a / s : 0.f;
}

} // namespace hippt
