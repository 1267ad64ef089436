// texture.h — bilinear texture views usable from host and device.
//
// Capability parity: reference src/core/textures.cuh (5 texture slots per
// BSDF: diffuse/specular/glossy/normal/roughness; lat-long envmap; TBN-frame
// normal mapping).  The reference rides NVIDIA's texture units
// (cudaTextureObject_t); gfx950/CDNA4 exposes NO device texture API (the
// HIP image intrinsics are compile-time unavailable for this arch), so the
// MI355X-native path is software bilinear filtering over RGBA32F rows in
// global memory — four 16-byte vector loads through L1/L2, which on CDNA is
// the fast path anyway (no texture cache exists).
#pragma once
#include "frame.h"

namespace hippt {

// Texture data is RGBA float4 rows (w*h*4 floats), wrap-repeat addressing.
struct TexView {
    unsigned long long tex_obj;  // reserved (0); kept for layout stability
    const float* data;           // RGBA32F pixels (device or host pointer)
    int w, h;
    int pad;

    HD bool valid() const { return data != nullptr; }

    HD Vec4 fetch(int x, int y) const {
        const float* p = data + 4 * (size_t(y) * w + x);
        return {p[0], p[1], p[2], p[3]};
    }

    HD Vec4 sample(Vec2 uv) const {
        // wrap repeat + bilinear on raw data
        float fx = uv.x - floorf(uv.x);
        float fy = uv.y - floorf(uv.y);
        float px = fx * w - 0.5f, py = fy * h - 0.5f;
        int x0 = (int)floorf(px), y0 = (int)floorf(py);
        float ax = px - x0, ay = py - y0;
        int x1 = x0 + 1, y1 = y0 + 1;
        x0 = (x0 % w + w) % w; x1 = (x1 % w + w) % w;
        y0 = (y0 % h + h) % h; y1 = (y1 % h + h) % h;
        Vec4 c00 = fetch(x0, y0), c10 = fetch(x1, y0), c01 = fetch(x0, y1), c11 = fetch(x1, y1);
        Vec4 a = c00 + (c10 - c00) * ax;
        Vec4 b = c01 + (c11 - c01) * ax;
        return a + (b - a) * ay;
    }
};

// 5 texture slots per BSDF (reference textures.cuh c_textures registry).
enum TexSlot : int { TEX_DIFFUSE = 0, TEX_SPECULAR = 1, TEX_GLOSSY = 2, TEX_NORMAL = 3, TEX_ROUGHNESS = 4, TEX_NSLOTS = 5 };

// Sample color slot with constant fallback.
HD Vec3 tex_or(const TexView* textures, int tex_id, Vec2 uv, const Vec3& fallback) {
    if (tex_id < 0) return fallback;
    Vec4 c = textures[tex_id].sample(uv);
    return c.xyz();
}

// Tangent-frame normal mapping (reference textures.cuh:79-88 eval_normal).
HD Vec3 apply_normal_map(const TexView* textures, int tex_id, Vec2 uv, const Vec3& shading_n) {
    if (tex_id < 0) return shading_n;
    Vec4 c = textures[tex_id].sample(uv);
    Vec3 tn = Vec3(c.x * 2.f - 1.f, c.y * 2.f - 1.f, c.z * 2.f - 1.f);
    float l2 = tn.length2();
    if (l2 < 1e-12f) return shading_n;
    return Frame::from_n(shading_n).to_world(tn * (1.f / sqrtf(l2))).normalized();
}

// Direction -> lat-long uv for environment maps.
HD Vec2 dir_to_latlong(const Vec3& d) {
    float u = 0.5f + atan2f(d.x, -d.z) * (0.5f / PI);
    float v = acosf(clampv(d.y, -1.f, 1.f)) * (1.f / PI);
    return {u, v};
}

} // namespace hippt
