// bind.cpp — pybind11 module `hippt._C`: scene assembly, CPU reference
// renderer entry points, GPU upload + kernel launches, BVH builders.
//
// Python (hippt/scene/scene.py) parses scenes and calls the setters here;
// the extension owns all device-side scene buffers (hipMalloc) while output
// images live in torch tensors whose data_ptr() is passed in for kernels to
// accumulate into (zero-copy handoff back to PyTorch-ROCm / RCCL).
#include <pybind11/pybind11.h>
#include <pybind11/numpy.h>
#include <pybind11/stl.h>
#include <cstring>
#include <stdexcept>

#include "core/scene_view.h"
#include "core/integrator.h"
#include "cpu/bvh_build.h"
#include "hip/kernels.h"

namespace py = pybind11;
using namespace hippt;

namespace hippt {
void render_cpu(const SceneView& sv, float* accum, float* var,
                int spp0, int nspp, uint32_t seed, int renderer, int n_threads,
                int y0, int y1, const uint8_t* spp_map, float* aux);
void render_lt_cpu(const SceneView& sv, float* accum, int spp0, int nspp, uint32_t seed,
                   int spec_constraint, float caustic_scaling, int n_threads);
}

namespace {

using farr = py::array_t<float, py::array::c_style | py::array::forcecast>;
using iarr = py::array_t<int32_t, py::array::c_style | py::array::forcecast>;
using uarr = py::array_t<uint32_t, py::array::c_style | py::array::forcecast>;

#define HIP_OK(expr)                                                                   \
    do {                                                                               \
        int _e = (expr);                                                               \
        if (_e != 0) throw std::runtime_error(std::string("HIP error ") +              \
                                              std::to_string(_e) + " at " #expr);      \
    } while (0)

Vec3 to_vec3(const std::vector<float>& v) { return {v[0], v[1], v[2]}; }

// Max level count of a 4-wide tree (host walk; used to gate halved-stack
// traversal kernels like the wavefront dual walk).
int bvh4_tree_depth(const BVH4Node* nodes, int n) {
    if (n <= 0) return 0;
    std::vector<std::pair<int, int>> stk{{0, 1}};
    int maxd = 1;
    while (!stk.empty()) {
        auto [idx, d] = stk.back();
        stk.pop_back();
        maxd = std::max(maxd, d);
        const BVH4Node& nd = nodes[idx];
        for (int c = 0; c < 4; ++c)
            if (nd.child[c] >= 0 && !(nd.child[c] == 0 && nd.cnt[c] == 0))
                stk.push_back({nd.child[c], d + 1});
    }
    return maxd;
}

struct SceneHolder {
    // ----- host data (kept alive / owned)
    farr np_prims, np_attrs, np_nodes, np_nodes4, np_nodes8;
    uarr np_prim_obj;
    iarr np_objs;
    std::vector<BsdfParams> bsdfs;
    std::vector<EmitterParams> emitters;
    std::vector<PhaseParams> phases;
    std::vector<MediumParams> media;          // host pointers inside
    std::vector<farr> media_density, media_temp;
    std::vector<std::vector<float>> media_super;   // majorant supergrids (raw max)
    std::vector<farr> tex_data;
    std::vector<TexView> tex_host;
    std::vector<int> emitter_prims;
    std::vector<float> emitter_cdf;
    std::vector<float> env_rows_h, env_cols_h;
    int env_w = 0, env_h = 0;
    std::vector<float> emitter_sel_h;
    Camera cam{};
    MaxDepthParams md{};
    int env_emitter = -1;
    int cam_medium = -1;

    // ----- device mirrors
    bool has_dev = false;
    int device_id = 0;
    std::vector<void*> dev_bufs;              // everything hipMalloc'ed
    std::vector<TexView> tex_dev;
    std::vector<MediumParams> media_dev;
    // offsets of updateable device arrays (hot reload)
    void* dev_bsdfs = nullptr;
    void* dev_emitters = nullptr;
    void* dev_media = nullptr;

    SceneView host_sv{};
    SceneView dev_sv{};
    WfState* wf = nullptr;

    ~SceneHolder() { release(); }

    void release() {
        if (wf) { wf_destroy(wf); wf = nullptr; }
        for (void* p : dev_bufs) dev_free(p);
        dev_bufs.clear();
        has_dev = false;
    }

    void set_geometry(farr prims, farr attrs, uarr prim_obj, farr nodes, farr nodes4,
                      farr nodes8) {
        if (prims.ndim() != 2 || prims.shape(1) != 12) throw std::runtime_error("prims must be (n,12)");
        if (attrs.ndim() != 2 || attrs.shape(1) != 16) throw std::runtime_error("attrs must be (n,16)");
        if (nodes.ndim() != 2 || nodes.shape(1) != 8) throw std::runtime_error("nodes must be (m,8)");
        if (nodes4.ndim() == 2 && nodes4.shape(0) > 0 && nodes4.shape(1) != 32)
            throw std::runtime_error("nodes4 must be (m,32)");
        np_prims = std::move(prims);
        np_attrs = std::move(attrs);
        np_prim_obj = std::move(prim_obj);
        np_nodes = std::move(nodes);
        np_nodes4 = std::move(nodes4);
        if (nodes8.ndim() == 2 && nodes8.shape(0) > 0 && nodes8.shape(1) != 64)
            throw std::runtime_error("nodes8 must be (m,64)");
        np_nodes8 = std::move(nodes8);
        bvh4_depth = np_nodes4.ndim() == 2 && np_nodes4.shape(0) > 0
            ? bvh4_tree_depth((const BVH4Node*)np_nodes4.data(), (int)np_nodes4.shape(0))
            : 0;
        nodes4q = np_nodes4.ndim() == 2 && np_nodes4.shape(0) > 0
            ? quantize_bvh4((const BVH4Node*)np_nodes4.data(), (int)np_nodes4.shape(0))
            : std::vector<BVH4NodeQ>();
        // scene bounding sphere (envmap sample_le disk origin)
        const Prim* pr = (const Prim*)np_prims.data();
        const uint32_t* po = np_prim_obj.data();
        int n = (int)np_prims.shape(0);
        AABB sb;
        for (int i = 0; i < n; ++i) {
            if (po[i] & PRIM_SPHERE_BIT) {
                Vec3 c = pr[i].v0.xyz();
                float rad = pr[i].v0.w;
                sb.grow(c - Vec3(rad)); sb.grow(c + Vec3(rad));
            } else {
                Vec3 v0 = pr[i].v0.xyz();
                sb.grow(v0);
                sb.grow(v0 + pr[i].e1.xyz());
                sb.grow(v0 + pr[i].e2.xyz());
            }
        }
        Vec3 c = (sb.lo + sb.hi) * 0.5f;
        float rad = n > 0 ? (sb.hi - sb.lo).length() * 0.5f : 1.f;
        scene_bound = Vec4(c, rad);
        tri_only = 1;
        for (int i = 0; i < n; ++i)
            if (po[i] & PRIM_SPHERE_BIT) { tri_only = 0; break; }
    }

    int bvh4_depth = 0;
    Vec4 scene_bound{0.f, 0.f, 0.f, 1.f};
    int cache_nodes = 0;
    int tri_only = 0;
    std::vector<BVH4NodeQ> nodes4q;   // quantized mirror of np_nodes4

    void set_cache_level(int level) {
        // accelerator XML cache_level -> top-tree nodes in LDS (2^level,
        // the reference's binary top-level count; clamped in the launcher
        // to the LDS budget)
        cache_nodes = level > 0 ? (1 << (level > 12 ? 12 : level)) : 0;
    }

    void set_objects(iarr objs) {
        if (objs.ndim() != 2 || objs.shape(1) != 8) throw std::runtime_error("objs must be (n,8)");
        np_objs = std::move(objs);
    }

    int add_bsdf(int type, std::vector<float> kd, std::vector<float> ks, std::vector<float> kg,
                 float ior, float extra0, float extra1, std::vector<int> tex) {
        BsdfParams b = BsdfParams::make(type);
        b.kd = Vec4(kd[0], kd[1], kd[2], kd.size() > 3 ? kd[3] : 0.f);
        b.ks = Vec4(ks[0], ks[1], ks[2], ks.size() > 3 ? ks[3] : 0.f);
        b.kg = Vec4(kg[0], kg[1], kg[2], kg.size() > 3 ? kg[3] : 0.f);
        b.ior = ior; b.extra0 = extra0; b.extra1 = extra1;
        for (size_t i = 0; i < tex.size() && i < 8; ++i) b.tex[i] = (int16_t)tex[i];
        bsdfs.push_back(b);
        return (int)bsdfs.size() - 1;
    }

    void update_bsdf(int i, int type, std::vector<float> kd, std::vector<float> ks,
                     std::vector<float> kg, float ior, float extra0, float extra1,
                     std::vector<int> tex) {
        if (i < 0 || i >= (int)bsdfs.size()) throw std::runtime_error("bad bsdf index");
        BsdfParams b = BsdfParams::make(type);
        b.kd = Vec4(kd[0], kd[1], kd[2], kd.size() > 3 ? kd[3] : 0.f);
        b.ks = Vec4(ks[0], ks[1], ks[2], ks.size() > 3 ? ks[3] : 0.f);
        b.kg = Vec4(kg[0], kg[1], kg[2], kg.size() > 3 ? kg[3] : 0.f);
        b.ior = ior; b.extra0 = extra0; b.extra1 = extra1;
        for (size_t k = 0; k < tex.size() && k < 8; ++k) b.tex[k] = (int16_t)tex[k];
        bsdfs[i] = b;
        if (has_dev && dev_bsdfs)
            HIP_OK(dev_upload((char*)dev_bsdfs + sizeof(BsdfParams) * i, &bsdfs[i], sizeof(BsdfParams)));
    }

    int add_emitter(int type, std::vector<float> emission, float scale, std::vector<float> aux,
                    int obj_id, int tex_id, int prim_base, int prim_cnt, float inv_area) {
        EmitterParams e{};
        e.emission = Vec4(emission[0], emission[1], emission[2], scale);
        e.aux = Vec4(aux[0], aux[1], aux[2], aux.size() > 3 ? aux[3] : 0.f);
        e.type = type; e.obj_id = obj_id; e.tex_id = tex_id;
        e.prim_base = prim_base; e.prim_cnt = prim_cnt; e.inv_area = inv_area;
        emitters.push_back(e);
        if (type == EM_ENVMAP) env_emitter = (int)emitters.size() - 1;
        return (int)emitters.size() - 1;
    }

    void update_emitter(int i, std::vector<float> emission, float scale, std::vector<float> aux) {
        if (i < 0 || i >= (int)emitters.size()) throw std::runtime_error("bad emitter index");
        emitters[i].emission = Vec4(emission[0], emission[1], emission[2], scale);
        emitters[i].aux = Vec4(aux[0], aux[1], aux[2], aux.size() > 3 ? aux[3] : 0.f);
        if (has_dev && dev_emitters)
            HIP_OK(dev_upload((char*)dev_emitters + sizeof(EmitterParams) * i, &emitters[i],
                              sizeof(EmitterParams)));
    }

    void set_emitter_prims(iarr eprims, farr ecdf) {
        emitter_prims.assign(eprims.data(), eprims.data() + eprims.size());
        emitter_cdf.assign(ecdf.data(), ecdf.data() + ecdf.size());
    }

    void set_emitter_sel(farr cdf) {
        emitter_sel_h.assign(cdf.data(), cdf.data() + cdf.size());
    }

    void set_env_cdf(farr rows, farr cols) {
        if (cols.ndim() != 2) throw std::runtime_error("env cols must be (h,w)");
        env_h = (int)cols.shape(0);
        env_w = (int)cols.shape(1);
        env_rows_h.assign(rows.data(), rows.data() + rows.size());
        env_cols_h.assign(cols.data(), cols.data() + cols.size());
        if ((int)env_rows_h.size() != env_h) throw std::runtime_error("env rows/cols mismatch");
    }

    int add_phase(int type, float g1, float g2, float wmix) {
        phases.push_back({type, g1, g2, wmix});
        return (int)phases.size() - 1;
    }

    int add_medium(int type, std::vector<float> sigma_a, std::vector<float> sigma_s,
                   int phase_id, std::vector<float> grid_lo, std::vector<float> grid_hi,
                   py::object density, py::object temperature,
                   float scale, float emission_scale, float temp_scale) {
        MediumParams m{};
        m.sigma_a = Vec4(sigma_a[0], sigma_a[1], sigma_a[2], 0.f);
        m.sigma_s = Vec4(sigma_s[0], sigma_s[1], sigma_s[2], 0.f);
        m.type = type; m.phase_id = phase_id;
        m.scale = scale; m.emission_scale = emission_scale; m.temp_scale = temp_scale;
        m.density = nullptr; m.temperature = nullptr;
        m.super = nullptr;
        m.nx = m.ny = m.nz = 0;
        m.sx = m.sy = m.sz = 0;
        if (type == MED_GRID) {
            farr d = density.cast<farr>();
            if (d.ndim() != 3) throw std::runtime_error("density must be (nz,ny,nx)");
            media_density.push_back(d);
            m.nz = (int)d.shape(0); m.ny = (int)d.shape(1); m.nx = (int)d.shape(2);
            m.density = media_density.back().data();
            float mx = 0.f; double sum = 0.0;
            const float* dd = m.density;
            size_t n = (size_t)m.nx * m.ny * m.nz;
            for (size_t i = 0; i < n; ++i) { mx = fmaxf(mx, dd[i]); sum += dd[i]; }
            m.majorant = mx * scale;
            m.avg_density = (float)(sum / std::max<size_t>(n, 1)) * scale;
            // majorant supergrid: per-8^3-supercell RAW density max with a
            // 1-voxel dilation (the stochastic-offset lookup samples up to
            // +-0.5 voxel outside the cell); scale is applied in the walk so
            // update_medium's scale changes need no rebuild
            // HIPPT_SUPER_N = voxels per supercell per axis (A/B: finer
            // cells skip empty space tighter but cost more DDA steps)
            static const int SUP = [] {
                const char* e = getenv("HIPPT_SUPER_N");
                int v = e ? atoi(e) : 16;  // smoke 1080p: sup4 74.4, sup8 84.9, sup16 89.6 Msps
                return v < 2 ? 2 : (v > 64 ? 64 : v);
            }();
            if (getenv("HIPPT_NO_SUPER")) {
                // A/B hook: single global majorant (round-1 behavior)
                m.sx = m.sy = m.sz = 0;
                m.super = nullptr;
                goto super_done;
            }
            m.sx = (m.nx + SUP - 1) / SUP;
            m.sy = (m.ny + SUP - 1) / SUP;
            m.sz = (m.nz + SUP - 1) / SUP;
            {
            std::vector<float> sup((size_t)m.sx * m.sy * m.sz, 0.f);
            for (int iz = 0; iz < m.nz; ++iz)
                for (int iy = 0; iy < m.ny; ++iy)
                    for (int ix = 0; ix < m.nx; ++ix) {
                        float v = dd[(size_t(iz) * m.ny + iy) * m.nx + ix];
                        if (v <= 0.f) continue;
                        int cx0 = std::max(0, (ix - 1) / SUP), cx1 = std::min(m.sx - 1, (ix + 1) / SUP);
                        int cy0 = std::max(0, (iy - 1) / SUP), cy1 = std::min(m.sy - 1, (iy + 1) / SUP);
                        int cz0 = std::max(0, (iz - 1) / SUP), cz1 = std::min(m.sz - 1, (iz + 1) / SUP);
                        for (int cz = cz0; cz <= cz1; ++cz)
                            for (int cy = cy0; cy <= cy1; ++cy)
                                for (int cx = cx0; cx <= cx1; ++cx) {
                                    float& s = sup[(size_t(cz) * m.sy + cy) * m.sx + cx];
                                    s = fmaxf(s, v);
                                }
                    }
            media_super.push_back(std::move(sup));
            m.super = media_super.back().data();
            }
            super_done:;
            Vec3 lo = to_vec3(grid_lo), hi = to_vec3(grid_hi);
            m.grid_lo = Vec4(lo, 0.f);
            Vec3 ext = hi - lo;
            m.grid_inv_extent = Vec4(1.f / ext.x, 1.f / ext.y, 1.f / ext.z, 0.f);
            if (!temperature.is_none()) {
                farr t = temperature.cast<farr>();
                media_temp.push_back(t);
                m.temperature = media_temp.back().data();
            }
        }
        media.push_back(m);
        return (int)media.size() - 1;
    }

    void update_medium(int i, std::vector<float> sigma_a, std::vector<float> sigma_s,
                       float scale, float emission_scale) {
        if (i < 0 || i >= (int)media.size()) throw std::runtime_error("bad medium index");
        MediumParams& m = media[i];
        float old_scale = m.scale;
        m.sigma_a = Vec4(sigma_a[0], sigma_a[1], sigma_a[2], 0.f);
        m.sigma_s = Vec4(sigma_s[0], sigma_s[1], sigma_s[2], 0.f);
        if (m.type == MED_GRID && old_scale > 0.f) {
            m.majorant *= scale / old_scale;
            m.avg_density *= scale / old_scale;
        }
        m.scale = scale; m.emission_scale = emission_scale;
        if (has_dev && dev_media) {
            MediumParams md2 = media_dev[i];
            md2.sigma_a = m.sigma_a; md2.sigma_s = m.sigma_s;
            md2.scale = m.scale; md2.majorant = m.majorant;
            md2.avg_density = m.avg_density; md2.emission_scale = m.emission_scale;
            media_dev[i] = md2;
            HIP_OK(dev_upload((char*)dev_media + sizeof(MediumParams) * i, &media_dev[i],
                              sizeof(MediumParams)));
        }
    }

    int add_texture(farr rgba) {
        if (rgba.ndim() != 3 || rgba.shape(2) != 4) throw std::runtime_error("texture must be (h,w,4)");
        tex_data.push_back(std::move(rgba));
        const farr& t = tex_data.back();
        TexView v{};
        v.tex_obj = 0;
        v.data = t.data();
        v.w = (int)t.shape(1);
        v.h = (int)t.shape(0);
        tex_host.push_back(v);
        return (int)tex_host.size() - 1;
    }

    void set_camera(std::vector<float> pos, std::vector<float> R_rows, float focal,
                    int w, int h, float aperture, float focal_dist, int ortho, float ortho_scale) {
        cam.pos = to_vec3(pos);
        cam.R = Mat3({R_rows[0], R_rows[1], R_rows[2]},
                     {R_rows[3], R_rows[4], R_rows[5]},
                     {R_rows[6], R_rows[7], R_rows[8]});
        cam.focal = focal; cam.w = w; cam.h = h;
        cam.aperture = aperture; cam.focal_dist = focal_dist;
        cam.ortho = ortho; cam.ortho_scale = ortho_scale;
    }

    void set_depths(int max_depth, int max_diffuse, int max_specular, int max_transmit,
                    int max_volume, float min_time, float max_time, int use_tof,
                    float radiance_clamp = 0.f) {
        md = {max_depth, max_diffuse, max_specular, max_transmit, max_volume,
              min_time, max_time, use_tof, radiance_clamp};
    }

    void fill_common(SceneView& sv) {
        sv.n_nodes = (int)np_nodes.shape(0) - 1;  // last row = sentinel
        sv.n_nodes4 = np_nodes4.ndim() == 2 ? (int)np_nodes4.shape(0) : 0;
        sv.n_nodes8 = np_nodes8.ndim() == 2 ? (int)np_nodes8.shape(0) : 0;
        sv.n_prims = (int)np_prims.shape(0);
        sv.n_objs = np_objs.ndim() == 2 ? (int)np_objs.shape(0) : 0;
        sv.n_bsdfs = (int)bsdfs.size();
        sv.n_emitters = (int)emitters.size();
        sv.n_textures = (int)tex_host.size();
        sv.n_media = (int)media.size();
        sv.env_emitter = env_emitter;
        sv.env_w = env_w;
        sv.env_h = env_h;
        sv.cam_medium = cam_medium;
        sv.cam = cam;
        sv.md = md;
        sv.bvh4_depth = bvh4_depth;
        sv.scene_bound = scene_bound;
        sv.cache_nodes = cache_nodes;
        sv.tri_only = tri_only;
    }

    void finalize() {
        fill_common(host_sv);
        host_sv.nodes = (const BVHNode*)np_nodes.data();
        host_sv.nodes4 = host_sv.n_nodes4 > 0 ? (const BVH4Node*)np_nodes4.data() : nullptr;
        host_sv.nodes4q = nodes4q.empty() ? nullptr : nodes4q.data();
        host_sv.nodes8 = host_sv.n_nodes8 > 0 ? (const BVH8Node*)np_nodes8.data() : nullptr;
        host_sv.prims = (const Prim*)np_prims.data();
        host_sv.attrs = (const PrimAttr*)np_attrs.data();
        host_sv.prim_obj = np_prim_obj.data();
        host_sv.objs = (const ObjInfo*)np_objs.data();
        host_sv.bsdfs = bsdfs.data();
        host_sv.emitters = emitters.data();
        host_sv.emitter_prims = emitter_prims.data();
        host_sv.emitter_cdf = emitter_cdf.data();
        host_sv.env_rows = env_rows_h.empty() ? nullptr : env_rows_h.data();
        host_sv.env_cols = env_cols_h.empty() ? nullptr : env_cols_h.data();
        host_sv.emitter_sel_cdf = emitter_sel_h.empty() ? nullptr : emitter_sel_h.data();
        host_sv.textures = tex_host.data();
        host_sv.media = media.data();
        host_sv.phases = phases.data();
    }

    template <typename T>
    T* upload_vec(const T* src, size_t count) {
        if (count == 0) return nullptr;
        void* p = nullptr;
        HIP_OK(dev_malloc(&p, sizeof(T) * count));
        dev_bufs.push_back(p);
        HIP_OK(dev_upload(p, src, sizeof(T) * count));
        return (T*)p;
    }

    void upload(int device) {
        release();
        device_id = device;
        HIP_OK(dev_set_device(device));
        finalize();
        fill_common(dev_sv);
        dev_sv.nodes = upload_vec((const BVHNode*)np_nodes.data(), np_nodes.shape(0));
        dev_sv.nodes4 = dev_sv.n_nodes4 > 0
            ? upload_vec((const BVH4Node*)np_nodes4.data(), np_nodes4.shape(0)) : nullptr;
        dev_sv.nodes4q = nodes4q.empty() ? nullptr
            : upload_vec(nodes4q.data(), nodes4q.size());
        dev_sv.nodes8 = dev_sv.n_nodes8 > 0
            ? upload_vec((const BVH8Node*)np_nodes8.data(), np_nodes8.shape(0)) : nullptr;
        dev_sv.prims = upload_vec((const Prim*)np_prims.data(), np_prims.shape(0));
        dev_sv.attrs = upload_vec((const PrimAttr*)np_attrs.data(), np_attrs.shape(0));
        dev_sv.prim_obj = upload_vec(np_prim_obj.data(), np_prim_obj.size());
        dev_sv.objs = upload_vec((const ObjInfo*)np_objs.data(), np_objs.shape(0));
        dev_sv.emitter_prims = upload_vec(emitter_prims.data(), emitter_prims.size());
        dev_sv.emitter_cdf = upload_vec(emitter_cdf.data(), emitter_cdf.size());
        dev_sv.env_rows = upload_vec(env_rows_h.data(), env_rows_h.size());
        dev_sv.env_cols = upload_vec(env_cols_h.data(), env_cols_h.size());
        dev_sv.emitter_sel_cdf = upload_vec(emitter_sel_h.data(), emitter_sel_h.size());
        // textures: RGBA32F rows in device global memory (gfx950 has no
        // device texture units — software bilinear is the CDNA-native path)
        tex_dev.clear();
        for (size_t i = 0; i < tex_host.size(); ++i) {
            TexView v = tex_host[i];
            v.data = upload_vec(tex_data[i].data(), (size_t)v.w * v.h * 4);
            tex_dev.push_back(v);
        }
        dev_sv.textures = upload_vec(tex_dev.data(), tex_dev.size());
        // media: re-point grids at device copies
        media_dev = media;
        for (size_t i = 0; i < media.size(); ++i) {
            if (media[i].density) {
                size_t n = (size_t)media[i].nx * media[i].ny * media[i].nz;
                media_dev[i].density = upload_vec(media[i].density, n);
            }
            if (media[i].temperature) {
                size_t n = (size_t)media[i].nx * media[i].ny * media[i].nz;
                media_dev[i].temperature = upload_vec(media[i].temperature, n);
            }
            if (media[i].super) {
                size_t n = (size_t)media[i].sx * media[i].sy * media[i].sz;
                media_dev[i].super = upload_vec(media[i].super, n);
            }
        }
        dev_sv.media = upload_vec(media_dev.data(), media_dev.size());
        dev_sv.phases = upload_vec(phases.data(), phases.size());
        dev_sv.bsdfs = upload_vec(bsdfs.data(), bsdfs.size());
        dev_sv.emitters = upload_vec(emitters.data(), emitters.size());
        dev_bsdfs = (void*)dev_sv.bsdfs;
        dev_emitters = (void*)dev_sv.emitters;
        dev_media = (void*)dev_sv.media;
        has_dev = true;
    }

    void render_host(farr accum, py::object var, int spp0, int nspp, uint32_t seed,
                     int renderer, int spec_constraint, float caustic_scaling, int n_threads,
                     int y0 = 0, int y1 = 0, py::object spp_map = py::none(),
                     py::object aux = py::none()) {
        finalize();
        float* vp = nullptr;
        farr var_arr;
        if (!var.is_none()) { var_arr = var.cast<farr>(); vp = var_arr.mutable_data(); }
        float* auxp = nullptr;
        farr aux_arr;
        if (!aux.is_none()) { aux_arr = aux.cast<farr>(); auxp = aux_arr.mutable_data(); }
        const uint8_t* smp = nullptr;
        py::array_t<uint8_t, py::array::c_style | py::array::forcecast> smp_arr;
        if (!spp_map.is_none()) {
            smp_arr = spp_map.cast<py::array_t<uint8_t, py::array::c_style | py::array::forcecast>>();
            smp = smp_arr.data();
        }
        py::gil_scoped_release rel;
        if (renderer == R_LIGHT_TRACE)
            render_lt_cpu(host_sv, accum.mutable_data(), spp0, nspp, seed,
                          spec_constraint, caustic_scaling, n_threads);
        else
            render_cpu(host_sv, accum.mutable_data(), vp, spp0, nspp, seed, renderer, n_threads,
                       y0, y1, smp, auxp);
    }

    void render_device(uintptr_t accum_ptr, uintptr_t var_ptr, int spp0, int nspp,
                       uint32_t seed, int renderer, int spec_constraint,
                       float caustic_scaling, uintptr_t stream, int y0 = 0, int y1 = 0,
                       uintptr_t spp_map_ptr = 0, uintptr_t aux_ptr = 0) {
        if (!has_dev) throw std::runtime_error("scene not uploaded to device");
        dev_sv.cam = cam;   // camera / depth params may have changed (hot reload)
        dev_sv.md = md;
        dev_sv.cam_medium = cam_medium;
        if (renderer == R_WAVEFRONT_PT) {
            if (!wf) {
                wf = wf_create(cam.w, cam.h);
                if (!wf) throw std::runtime_error("wavefront state allocation failed");
            }
            py::gil_scoped_release rel;
            HIP_OK(launch_render_wavefront(wf, dev_sv, (float*)accum_ptr, (float*)var_ptr,
                                           spp0, nspp, seed, 0, (void*)stream));
            return;
        }
        HIP_OK(launch_render(dev_sv, (float*)accum_ptr, (float*)var_ptr, spp0, nspp, seed,
                             renderer, spec_constraint, caustic_scaling, (void*)stream, y0, y1,
                             (const uint8_t*)spp_map_ptr, (float*)aux_ptr));
    }

    py::dict info() {
        py::dict d;
        d["n_prims"] = np_prims.ndim() == 2 ? (int)np_prims.shape(0) : 0;
        d["n_nodes"] = np_nodes.ndim() == 2 ? (int)np_nodes.shape(0) : 0;
        d["n_bsdfs"] = (int)bsdfs.size();
        d["n_emitters"] = (int)emitters.size();
        d["n_media"] = (int)media.size();
        d["n_textures"] = (int)tex_host.size();
        d["has_dev"] = has_dev;
        return d;
    }
};

// ------------------------------------------------------------- BVH builder
py::tuple py_build_bvh(farr prims, uarr prim_obj, int max_leaf, float overlap_w,
                       bool use_sbvh, bool ref_unsplit, float trav_cost) {
    int n = (int)prims.shape(0);
    BVHBuildConfig cfg;
    cfg.max_leaf_prims = max_leaf;
    cfg.overlap_w = overlap_w;
    cfg.use_sbvh = use_sbvh;
    cfg.ref_unsplit = ref_unsplit;
    cfg.trav_cost = trav_cost;
    BVHBuildResult res;
    {
        py::gil_scoped_release rel;
        res = use_sbvh ? build_sbvh((const Prim*)prims.data(), prim_obj.data(), n, cfg)
                       : build_bvh((const Prim*)prims.data(), prim_obj.data(), n, cfg);
    }
    farr nodes({(py::ssize_t)res.nodes.size(), (py::ssize_t)8});
    std::memcpy(nodes.mutable_data(), res.nodes.data(), res.nodes.size() * sizeof(BVHNode));
    iarr order((py::ssize_t)res.prim_order.size());
    std::memcpy(order.mutable_data(), res.prim_order.data(), res.prim_order.size() * sizeof(int));
    py::dict stats;
    stats["n_leaves"] = res.n_leaves;
    stats["max_depth"] = res.max_depth;
    stats["sah_cost"] = res.sah_cost;
    return py::make_tuple(nodes, order, stats);
}

py::tuple py_collapse_bvh8(farr nodes) {
    if (nodes.ndim() != 2 || nodes.shape(1) != 8) throw std::runtime_error("nodes must be (m,8)");
    std::vector<BVHNode> bin((size_t)nodes.shape(0));
    std::memcpy(bin.data(), nodes.data(), bin.size() * sizeof(BVHNode));
    int depth8 = 0;
    std::vector<BVH8Node> n8;
    {
        py::gil_scoped_release rel;
        n8 = collapse_bvh8(bin, &depth8);
    }
    if (7 * depth8 > BVH4_STACK)
        throw std::runtime_error("BVH8 depth exceeds traversal stack bound");
    farr out({(py::ssize_t)n8.size(), (py::ssize_t)64});
    std::memcpy(out.mutable_data(), n8.data(), n8.size() * sizeof(BVH8Node));
    return py::make_tuple(out, depth8);
}

// Collapse a (m,8) skip-link node array into a (k,32) BVH4 array (bvh4.h).
py::tuple py_collapse_bvh4(farr nodes) {
    if (nodes.ndim() != 2 || nodes.shape(1) != 8) throw std::runtime_error("nodes must be (m,8)");
    std::vector<BVHNode> bin((size_t)nodes.shape(0));
    std::memcpy(bin.data(), nodes.data(), bin.size() * sizeof(BVHNode));
    int depth4 = 0;
    std::vector<BVH4Node> n4;
    {
        py::gil_scoped_release rel;
        n4 = collapse_bvh4(bin, &depth4);
    }
    if (3 * depth4 > BVH4_STACK)
        throw std::runtime_error("BVH4 depth exceeds traversal stack bound");
    farr out({(py::ssize_t)n4.size(), (py::ssize_t)32});
    std::memcpy(out.mutable_data(), n4.data(), n4.size() * sizeof(BVH4Node));
    return py::make_tuple(out, depth4);
}

// Traversal self-test: compare binary skip-link vs 4-wide ordered traversal
// (closest hit t/prim and occlusion verdict) on caller-supplied rays.
// Returns the number of mismatching rays.
int py_bvh4_selftest(farr prims, uarr prim_obj, farr nodes, farr nodes4,
                     farr ray_o, farr ray_d, float tmax, farr nodes8) {
    const BVH8Node* n8p = (nodes8.ndim() == 2 && nodes8.shape(0) > 0)
        ? (const BVH8Node*)nodes8.data() : nullptr;
    int n_nodes = (int)nodes.shape(0);
    const BVHNode* bn = (const BVHNode*)nodes.data();
    const BVH4Node* n4 = (const BVH4Node*)nodes4.data();
    const Prim* pr = (const Prim*)prims.data();
    const uint32_t* po = prim_obj.data();
    int bad = 0;
    int nr = (int)ray_o.shape(0);
    for (int i = 0; i < nr; ++i) {
        Ray r;
        r.o = {ray_o.at(i, 0), ray_o.at(i, 1), ray_o.at(i, 2)};
        r.d = {ray_d.at(i, 0), ray_d.at(i, 1), ray_d.at(i, 2)};
        HitRecord a = ray_intersect_bvh(bn, n_nodes, pr, po, r, tmax);
        HitRecord b = ray_intersect_bvh4(n4, pr, po, r, tmax);
        HitRecord w = ray_intersect_bvh4_ww(n4, pr, po, r, tmax);
        HitRecord w32 = ray_intersect_bvh4_ww32(n4, pr, po, r, tmax);
        if ((w32.prim_idx < 0) != (b.prim_idx < 0) ||
            (b.prim_idx >= 0 && fabsf(w32.t - b.t) > 1e-5f * fmaxf(1.f, b.t))) { ++bad; continue; }
        if ((w.prim_idx < 0) != (b.prim_idx < 0) ||
            (b.prim_idx >= 0 && fabsf(w.t - b.t) > 1e-5f * fmaxf(1.f, b.t))) { ++bad; continue; }
        {
            Bvh4Walk wk;
            bvh4_walk_init(wk, r, tmax);
            while (bvh4_walk_step(wk, n4, pr, po, nullptr, 0)) {}
            if (wk.rec.prim_idx < 0) wk.rec.t = MAX_DIST;
            if ((wk.rec.prim_idx < 0) != (b.prim_idx < 0) ||
                (b.prim_idx >= 0 && fabsf(wk.rec.t - b.t) > 1e-5f * fmaxf(1.f, b.t))) { ++bad; continue; }
        }
        if (n8p) {
            HitRecord w8 = ray_intersect_bvh8_ww(n8p, pr, po, r, tmax);
            bool occ8 = occlusion_test_bvh8(n8p, pr, po, r, tmax);
            bool occ_ref = occlusion_test_bvh4(n4, pr, po, r, tmax);
            if ((w8.prim_idx < 0) != (b.prim_idx < 0) ||
                (b.prim_idx >= 0 && fabsf(w8.t - b.t) > 1e-5f * fmaxf(1.f, b.t)) ||
                occ8 != occ_ref) { ++bad; continue; }
        }
        bool occ_a = occlusion_test_bvh(bn, n_nodes, pr, po, r, tmax);
        bool occ_b = occlusion_test_bvh4(n4, pr, po, r, tmax);
        if (occlusion_test_bvh4_ww(n4, pr, po, r, tmax) != occ_b) { ++bad; continue; }
        // prim index may differ only on exact t ties; compare t and object
        bool hit_match = (a.prim_idx < 0) == (b.prim_idx < 0) &&
                         (a.prim_idx < 0 || fabsf(a.t - b.t) <= 1e-5f * fmaxf(1.f, a.t));
        if (!hit_match || occ_a != occ_b) ++bad;
    }
    return bad;
}

// Quantized-walk self-test: the Q tree must agree with the fp32 walk on
// hit/miss, occlusion, and hit distance (quantized boxes are conservative,
// so the same closest prim must be found).  Returns mismatch count.
int py_bvh4q_selftest(farr prims, uarr prim_obj, farr nodes4,
                      farr ray_o, farr ray_d, float tmax) {
    const BVH4Node* n4 = (const BVH4Node*)nodes4.data();
    int nn = (int)nodes4.shape(0);
    std::vector<BVH4NodeQ> q = quantize_bvh4(n4, nn);
    const Prim* pr = (const Prim*)prims.data();
    const uint32_t* po = prim_obj.data();
    int bad = 0;
    int nr = (int)ray_o.shape(0);
    for (int i = 0; i < nr; ++i) {
        Ray r;
        r.o = {ray_o.at(i, 0), ray_o.at(i, 1), ray_o.at(i, 2)};
        r.d = {ray_d.at(i, 0), ray_d.at(i, 1), ray_d.at(i, 2)};
        HitRecord a = ray_intersect_bvh4_ww(n4, pr, po, r, tmax);
        HitRecord b = ray_intersect_bvh4q_ww(q.data(), pr, po, r, tmax);
        if ((a.prim_idx < 0) != (b.prim_idx < 0) ||
            (a.prim_idx >= 0 && fabsf(a.t - b.t) > 1e-5f * fmaxf(1.f, a.t))) { ++bad; continue; }
        bool oa = occlusion_test_bvh4_ww(n4, pr, po, r, tmax);
        bool ob = occlusion_test_bvh4q_ww(q.data(), pr, po, r, tmax);
        if (oa != ob) ++bad;
    }
    return bad;
}

// Closest-hit query over the BVH4 tree: returns (t, prim) arrays for rays.
py::tuple py_bvh4_hit(farr prims, uarr prim_obj, farr nodes4,
                      farr ray_o, farr ray_d) {
    const BVH4Node* n4 = (const BVH4Node*)nodes4.data();
    const Prim* pr = (const Prim*)prims.data();
    const uint32_t* po = prim_obj.data();
    int nr = (int)ray_o.shape(0);
    farr t_out(nr);
    iarr p_out(nr);
    for (int i = 0; i < nr; ++i) {
        Ray r;
        r.o = {ray_o.at(i, 0), ray_o.at(i, 1), ray_o.at(i, 2)};
        r.d = {ray_d.at(i, 0), ray_d.at(i, 1), ray_d.at(i, 2)};
        HitRecord h = ray_intersect_bvh4_ww(n4, pr, po, r, MAX_DIST);
        t_out.mutable_at(i) = h.t;
        p_out.mutable_at(i) = h.prim_idx;
    }
    return py::make_tuple(t_out, p_out);
}

// BSDF triple-consistency check on bsdf index i of the holder's table:
//  A = E_sample[weight]                 (hemispherical reflectance via sample)
//  B = 2pi * E_uniform[eval]            (same integral via eval, uniform dirs)
//  C = 2pi * E_uniform[pdf]             (pdf normalization, ~1 for non-delta)
// Mutually consistent sample/eval/pdf must give A ~= B; C ~= (lobe coverage).
py::tuple py_bsdf_check(SceneHolder& h, int i, float cos_o, uint32_t seed, int n) {
    if (i < 0 || i >= (int)h.bsdfs.size()) throw std::runtime_error("bad bsdf index");
    h.finalize();
    const BsdfParams& b = h.bsdfs[i];
    Interaction it{};
    it.shading_n = Vec3(0.f, 0.f, 1.f);
    float so = sqrtf(fmaxf(0.f, 1.f - cos_o * cos_o));
    Vec3 wo(so, 0.f, cos_o);
    Sampler sp(911u, seed);
    Vec3 A(0.f), B(0.f);
    double Cp = 0.0;
    for (int k = 0; k < n; ++k) {
        BsdfSample s = bsdf_sample(b, wo, it, sp, h.host_sv.textures);
        if (s.pdf > 0.f) A += s.weight;
        // uniform hemisphere direction
        Vec2 u = sp.next2f();
        float z = u.x, r = sqrtf(fmaxf(0.f, 1.f - z * z)), phi = 2.f * PI * u.y;
        Vec3 wi(r * cosf(phi), r * sinf(phi), z);
        B += bsdf_eval(b, wo, wi, it, h.host_sv.textures);
        Cp += (double)bsdf_pdf(b, wo, wi, it, h.host_sv.textures);
    }
    float inv_n = 1.f / (float)n;
    Vec3 a = A * inv_n;
    Vec3 bb = B * (2.f * PI * inv_n);
    double c = Cp * 2.0 * PI * inv_n;
    return py::make_tuple(py::make_tuple(a.x, a.y, a.z),
                          py::make_tuple(bb.x, bb.y, bb.z), c);
}

// Envmap NEE integral check: A = E[radiance/pdf] over emitter_sample draws
// (hemisphere above n=(0,0,1)) must equal B = the true integral of
// envmap radiance over that hemisphere (uniform-direction reference).
// Tight validation of the luminance-CDF sampler's pdf normalization.
py::tuple py_env_check(SceneHolder& h, uint32_t seed, int n) {
    h.finalize();
    if (h.env_emitter < 0) throw std::runtime_error("no envmap emitter");
    const EmitterParams& e = h.emitters[h.env_emitter];
    EmitterGeom g = h.host_sv.emitter_geom();
    Sampler sp(417u, seed);
    Vec3 A(0.f), B(0.f);
    Vec3 sp_n(0.f, 0.f, 1.f), sp_pos(0.f);
    for (int k = 0; k < n; ++k) {
        EmitterSampleRec r = emitter_sample(e, g, sp_pos, sp_n, sp);
        if (r.pdf > 0.f) {
            Vec3 dir = (r.pos - sp_pos).normalized();
            if (dir.z > 0.f) A += r.radiance / r.pdf;   // hemisphere only
        }
        Vec2 u = sp.next2f();
        float z = u.x, rr = sqrtf(fmaxf(0.f, 1.f - z * z)), phi = 2.f * PI * u.y;
        Vec3 wi(rr * cosf(phi), rr * sinf(phi), z);
        B += envmap_eval(e, wi, h.host_sv.textures);
    }
    float inv_n = 1.f / (float)n;
    Vec3 a = A * inv_n;
    Vec3 b = B * (2.f * PI * inv_n);
    return py::make_tuple(py::make_tuple(a.x, a.y, a.z),
                          py::make_tuple(b.x, b.y, b.z));
}

} // namespace

PYBIND11_MODULE(_C, m) {
    m.doc() = "hippt native core: MI355X path tracing kernels + scene runtime";

    py::class_<SceneHolder>(m, "Scene")
        .def(py::init<>())
        .def("set_geometry", &SceneHolder::set_geometry)
        .def("set_objects", &SceneHolder::set_objects)
        .def("add_bsdf", &SceneHolder::add_bsdf)
        .def("update_bsdf", &SceneHolder::update_bsdf)
        .def("add_emitter", &SceneHolder::add_emitter)
        .def("update_emitter", &SceneHolder::update_emitter)
        .def("set_emitter_prims", &SceneHolder::set_emitter_prims)
        .def("set_env_cdf", &SceneHolder::set_env_cdf)
        .def("set_emitter_sel", &SceneHolder::set_emitter_sel)
        .def("add_phase", &SceneHolder::add_phase)
        .def("add_medium", &SceneHolder::add_medium)
        .def("update_medium", &SceneHolder::update_medium)
        .def("add_texture", &SceneHolder::add_texture)
        .def("set_camera", &SceneHolder::set_camera)
        .def("set_depths", &SceneHolder::set_depths,
             py::arg("max_depth"), py::arg("max_diffuse"), py::arg("max_specular"),
             py::arg("max_transmit"), py::arg("max_volume"), py::arg("min_time"),
             py::arg("max_time"), py::arg("use_tof"), py::arg("radiance_clamp") = 0.f)
        .def("set_cache_level", &SceneHolder::set_cache_level)
        .def_readwrite("cam_medium", &SceneHolder::cam_medium)
        .def_readwrite("env_emitter", &SceneHolder::env_emitter)
        .def("finalize", &SceneHolder::finalize)
        .def("upload", &SceneHolder::upload)
        .def("release", &SceneHolder::release)
        .def("render_host", &SceneHolder::render_host,
             py::arg("accum"), py::arg("var"), py::arg("spp0"), py::arg("nspp"),
             py::arg("seed"), py::arg("renderer"), py::arg("spec_constraint"),
             py::arg("caustic_scaling"), py::arg("n_threads"),
             py::arg("y0") = 0, py::arg("y1") = 0, py::arg("spp_map") = py::none(),
             py::arg("aux") = py::none())
        .def("render_device", &SceneHolder::render_device,
             py::arg("accum_ptr"), py::arg("var_ptr"), py::arg("spp0"), py::arg("nspp"),
             py::arg("seed"), py::arg("renderer"), py::arg("spec_constraint"),
             py::arg("caustic_scaling"), py::arg("stream"),
             py::arg("y0") = 0, py::arg("y1") = 0, py::arg("spp_map_ptr") = 0,
             py::arg("aux_ptr") = 0)
        .def("info", &SceneHolder::info);

    m.def("build_bvh", &py_build_bvh, py::arg("prims"), py::arg("prim_obj"),
          py::arg("max_leaf") = 4, py::arg("overlap_w") = 0.f,
          py::arg("use_sbvh") = false, py::arg("ref_unsplit") = true,
          py::arg("trav_cost") = 0.f);
    m.def("collapse_bvh4", &py_collapse_bvh4, py::arg("nodes"));
    m.def("collapse_bvh8", &py_collapse_bvh8, py::arg("nodes"));
    m.def("bvh4_hit", &py_bvh4_hit);
    m.def("bsdf_check", &py_bsdf_check);
    m.def("env_check", &py_env_check);
    m.def("bvh4q_selftest", &py_bvh4q_selftest,
          py::arg("prims"), py::arg("prim_obj"), py::arg("nodes4"),
          py::arg("ray_o"), py::arg("ray_d"), py::arg("tmax"));
    m.def("bvh4_selftest", &py_bvh4_selftest,
          py::arg("prims"), py::arg("prim_obj"), py::arg("nodes"), py::arg("nodes4"),
          py::arg("ray_o"), py::arg("ray_d"), py::arg("tmax"),
          py::arg("nodes8") = farr());

    m.def("dev_synchronize", [] { HIP_OK(dev_synchronize()); });
    m.def("dev_set_device", [](int d) { HIP_OK(dev_set_device(d)); });

    m.def("struct_sizes", [] {
        py::dict d;
        d["BVHNode"] = (int)sizeof(BVHNode);
        d["Prim"] = (int)sizeof(Prim);
        d["PrimAttr"] = (int)sizeof(PrimAttr);
        d["ObjInfo"] = (int)sizeof(ObjInfo);
        d["BsdfParams"] = (int)sizeof(BsdfParams);
        d["EmitterParams"] = (int)sizeof(EmitterParams);
        d["MediumParams"] = (int)sizeof(MediumParams);
        d["PhaseParams"] = (int)sizeof(PhaseParams);
        d["TexView"] = (int)sizeof(TexView);
        d["Camera"] = (int)sizeof(Camera);
        d["SceneView"] = (int)sizeof(SceneView);
        return d;
    });

    m.attr("R_MEGAKERNEL_PT") = (int)R_MEGAKERNEL_PT;
    m.attr("R_WAVEFRONT_PT") = (int)R_WAVEFRONT_PT;
    m.attr("R_VOLUME_PT") = (int)R_VOLUME_PT;
    m.attr("R_LIGHT_TRACE") = (int)R_LIGHT_TRACE;
    m.attr("R_DEPTH") = (int)R_DEPTH;
    m.attr("R_BVH_COST") = (int)R_BVH_COST;
    m.attr("R_MEGAKERNEL_PT_DYN") = (int)R_MEGAKERNEL_PT_DYN;
    m.attr("WAVE_SIZE") = 64;
}
