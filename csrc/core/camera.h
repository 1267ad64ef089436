// camera.h — on-device camera model: pinhole / orthographic / thin-lens DoF.
//
// Capability parity: reference src/core/camera_model.cuh (device ray-gen with
// pixel jitter, DoF lens sampling, ortho branch; get_splat_pixel inverse
// projection for light tracing) + impl/camera_model.cu (fov<->focal, WASD
// move, yaw/pitch rotate — host side lives in hippt/scene/camera.py).
#pragma once
#include "frame.h"
#include "geometry.h"
#include "sampling.h"
#include "rng.h"

namespace hippt {

struct alignas(16) Camera {
    Mat3 R;            // camera->world rotation; columns: right, up, forward
    Vec3 pos;
    float focal;       // focal length in pixels: 0.5*w / tan(0.5*fov_x)
    int w, h;
    float aperture;    // lens radius (0 = pinhole)
    float focal_dist;  // focus distance for DoF
    int ortho;         // orthographic camera flag
    float ortho_scale; // world units per pixel for ortho
    float pad0, pad1;

    // Generate primary ray through pixel (px, py) with jitter.
    // samp >= 0 stratifies the AA jitter over a per-pixel-rotated 4x4 grid
    // cycling with the sample index (extension; same RNG stream consumption,
    // so determinism and all downstream sampling are unchanged)
    HD Ray gen_ray(int px, int py, Sampler& sp, int samp = -1) const {
        Vec2 j = sp.next2f();
        if (samp >= 0) {
            uint32_t hh = (uint32_t)(px * 9781 + py * 6271);
            uint32_t cell = ((uint32_t)samp + (hh ^ (hh >> 16))) & 15u;
            j = Vec2(((float)(cell & 3u) + j.x) * 0.25f,
                     ((float)(cell >> 2) + j.y) * 0.25f);
        }
        float x = (px + j.x - 0.5f * w);
        float y = (0.5f * h - py - j.y);
        if (ortho) {
            Vec3 o = pos + R * Vec3(x * ortho_scale, y * ortho_scale, 0.f);
            return Ray(o, R * Vec3(0.f, 0.f, 1.f));
        }
        Vec3 d_cam = Vec3(x, y, focal).normalized();
        if (aperture > 0.f) {
            // thin-lens: jitter origin on the aperture disk, focus at focal_dist
            Vec2 lens = sample_concentric_disk(sp.next2f()) * aperture;
            float ft = focal_dist / d_cam.z;
            Vec3 focus = d_cam * ft;
            Vec3 o_cam = Vec3(lens.x, lens.y, 0.f);
            Vec3 nd = (focus - o_cam).normalized();
            return Ray(pos + R * o_cam, R * nd);
        }
        return Ray(pos, R * d_cam);
    }

    // Inverse projection for light-tracing splats (camera_model.cuh:92-104).
    // Returns pixel coords; valid=false if behind camera / outside frame.
    HD bool get_splat_pixel(const Vec3& world_pos, int& px, int& py) const {
        Vec3 p_cam = R.t_mul(world_pos - pos);
        if (ortho) {
            px = (int)floorf(p_cam.x / ortho_scale + 0.5f * w);
            py = (int)floorf(0.5f * h - p_cam.y / ortho_scale);
        } else {
            if (p_cam.z <= 1e-5f) return false;
            float inv_z = 1.f / p_cam.z;
            px = (int)floorf(p_cam.x * focal * inv_z + 0.5f * w);
            py = (int)floorf(0.5f * h - p_cam.y * focal * inv_z);
        }
        return px >= 0 && px < w && py >= 0 && py < h;
    }
};

} // namespace hippt
