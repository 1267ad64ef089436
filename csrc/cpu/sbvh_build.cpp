// sbvh_build.cpp — spatial-split BVH (SBVH, Stich et al. 2009 style).
//
// Capability parity: reference src/impl/bvh_spatial.cu (object-vs-spatial
// split choice per node, SpatialSplitter chopped binning with exact
// triangle-AABB clipping :278-425, reference unsplitting :468-529,
// multithreaded build) + src/impl/proc_geometry.cu (Sutherland-Hodgman
// clipping).  Our build parallelizes the top levels with std::thread forks
// instead of the reference's 8-thread pool + lock-free work stealing — same
// effect (top splits dominate), much less machinery.
//
// Output contract matches build_bvh: DFS skip-link nodes + prim_order, where
// prim_order may contain DUPLICATE original indices (spatial splits create
// multiple references to one triangle); the Python layer gathers primitive
// arrays through prim_order so duplication is transparent to traversal.
#include "bvh_build.h"
#include <algorithm>
#include <cmath>
#include <thread>
#include <vector>

namespace hippt {

namespace {

constexpr int NB = 16;          // bins (both split kinds)
constexpr float ALPHA = 1e-5f;  // overlap threshold that triggers spatial-split search

struct Refr {
    AABB box;
    int prim;
};

struct SNode {
    AABB box;
    int left = -1, right = -1;
    int leaf_base = 0, leaf_cnt = 0;   // into the final ref list
};

struct Ctx {
    const Prim* prims;
    const uint32_t* prim_obj;
    int max_leaf;
    bool ref_unsplit;
    float inv_root_area;
    float trav_cost = 0.f;
    // final leaves' reference order (append-only under mutex-free ownership:
    // built single-threaded per subtree, merged bottom-up)
};

// ---- Sutherland-Hodgman clip of a convex polygon against axis slab plane
// keep points with (p[axis] >= bound) if lower, else (p[axis] <= bound).
int clip_plane(const Vec3* in, int n, Vec3* out, int axis, float bound, bool lower) {
    int m = 0;
    for (int i = 0; i < n; ++i) {
        const Vec3& a = in[i];
        const Vec3& b = in[(i + 1) % n];
        float da = lower ? a[axis] - bound : bound - a[axis];
        float db = lower ? b[axis] - bound : bound - b[axis];
        bool ina = da >= 0.f, inb = db >= 0.f;
        if (ina) out[m++] = a;
        if (ina != inb) {
            float t = da / (da - db);
            out[m++] = a + (b - a) * t;
        }
    }
    return m;
}

// AABB of triangle clipped to [lo, hi] along `axis` (other axes unclipped),
// intersected with the reference's current box.
AABB clip_tri_to_slab(const Prim& p, bool is_sphere, const AABB& refbox,
                      int axis, float lo, float hi) {
    if (is_sphere) {
        AABB b = refbox;
        b.lo.set(axis, fmaxf(b.lo[axis], lo));
        b.hi.set(axis, fminf(b.hi[axis], hi));
        return b;
    }
    Vec3 poly[8], tmp[8];
    poly[0] = p.v0.xyz();
    poly[1] = p.v0.xyz() + p.e1.xyz();
    poly[2] = p.v0.xyz() + p.e2.xyz();
    int n = 3;
    n = clip_plane(poly, n, tmp, axis, lo, true);
    n = clip_plane(tmp, n, poly, axis, hi, false);
    AABB b;
    for (int i = 0; i < n; ++i) b.grow(poly[i]);
    // intersect with the (possibly already clipped) reference box
    b.lo = b.lo.maxv(refbox.lo);
    b.hi = b.hi.minv(refbox.hi);
    return b;
}

struct BuildOut {
    std::vector<SNode> nodes;      // local pool, root at index 0
    std::vector<Refr> leaf_refs;   // refs in leaf order
    int n_leaves = 0;
    int max_depth = 0;
    int n_spatial = 0;
};

// merge child outputs into parent
int merge_child(BuildOut& dst, const BuildOut& src) {
    int node_off = (int)dst.nodes.size();
    int ref_off = (int)dst.leaf_refs.size();
    for (SNode nd : src.nodes) {
        if (nd.left >= 0) { nd.left += node_off; nd.right += node_off; }
        else nd.leaf_base += ref_off;
        dst.nodes.push_back(nd);
    }
    dst.leaf_refs.insert(dst.leaf_refs.end(), src.leaf_refs.begin(), src.leaf_refs.end());
    dst.n_leaves += src.n_leaves;
    dst.max_depth = std::max(dst.max_depth, src.max_depth);
    dst.n_spatial += src.n_spatial;
    return node_off;
}

void build_rec(const Ctx& ctx, std::vector<Refr>&& refs, int depth, BuildOut& out) {
    AABB box;
    for (const Refr& r : refs) box.grow(r.box);
    int n = (int)refs.size();
    int my = (int)out.nodes.size();
    out.nodes.emplace_back();
    out.nodes[my].box = box;
    out.max_depth = std::max(out.max_depth, depth);

    auto make_leaf = [&] {
        SNode& nd = out.nodes[my];
        nd.leaf_base = (int)out.leaf_refs.size();
        nd.leaf_cnt = n;
        out.leaf_refs.insert(out.leaf_refs.end(), refs.begin(), refs.end());
        ++out.n_leaves;
    };
    // Forced median split for every fallback: the ww traversal packs the
    // leaf prim count into 4 bits (bvh4.h, cnt<<27), so a leaf may hold at
    // most 15 refs.  A median split always halves n, so recursion from any
    // fallback site terminates with all leaves <= 15.
    auto median_split = [&] {
        std::vector<Refr> l(refs.begin(), refs.begin() + n / 2);
        std::vector<Refr> r(refs.begin() + n / 2, refs.end());
        refs.clear(); refs.shrink_to_fit();
        build_rec(ctx, std::move(l), depth + 1, out);
        int li = my + 1;
        int ri = (int)out.nodes.size();
        build_rec(ctx, std::move(r), depth + 1, out);
        out.nodes[my].left = li; out.nodes[my].right = ri;
    };
    if (n <= 1 || (depth > 60 && n <= 15)) { make_leaf(); return; }
    if (depth > 60) { median_split(); return; }

    const float leaf_cost = box.area() * n;

    // ---------- object split (binned SAH over centroids)
    AABB cbox;
    for (const Refr& r : refs) cbox.grow(r.box.centroid());
    Vec3 cext = cbox.extent();
    int best_obj_axis = -1, best_obj_bin = -1;
    float best_obj_cost = 1e30f;
    AABB best_obj_lbox, best_obj_rbox;
    for (int axis = 0; axis < 3; ++axis) {
        float ext = cext[axis];
        if (ext < 1e-12f) continue;
        float cmin = cbox.lo[axis];
        float inv = NB / ext;
        AABB bins[NB]; int cnt[NB] = {0};
        for (const Refr& r : refs) {
            int b = std::min(NB - 1, (int)((r.box.centroid()[axis] - cmin) * inv));
            bins[b].grow(r.box); ++cnt[b];
        }
        AABB lb[NB]; int lc[NB];
        AABB acc; int c = 0;
        for (int b = 0; b < NB; ++b) { acc.grow(bins[b]); c += cnt[b]; lb[b] = acc; lc[b] = c; }
        AABB rb[NB]; AABB racc;
        for (int b = NB - 1; b >= 1; --b) { racc.grow(bins[b]); rb[b] = racc; }
        for (int b = 0; b < NB - 1; ++b) {
            int nl = lc[b], nr = n - nl;
            if (nl == 0 || nr == 0) continue;
            float cost = lb[b].area() * nl + rb[b + 1].area() * nr;
            if (cost < best_obj_cost) {
                best_obj_cost = cost; best_obj_axis = axis; best_obj_bin = b;
                best_obj_lbox = lb[b]; best_obj_rbox = rb[b + 1];
            }
        }
    }

    // ---------- spatial split (chopped binning), tried when object-split
    // children overlap significantly (Stich et al. alpha test)
    int best_sp_axis = -1, best_sp_bin = -1;
    float best_sp_cost = 1e30f;
    float lambda = best_obj_axis >= 0
                       ? AABB::intersection_area(best_obj_lbox, best_obj_rbox) : 0.f;
    bool try_spatial = best_obj_axis >= 0 &&
                       lambda * ctx.inv_root_area > ALPHA;
    Vec3 ext = box.extent();
    if (try_spatial) {
        for (int axis = 0; axis < 3; ++axis) {
            float e = ext[axis];
            if (e < 1e-10f) continue;
            float lo = box.lo[axis];
            float width = e / NB;
            float inv = 1.f / width;
            AABB bins[NB]; int entry[NB] = {0}, exit_[NB] = {0};
            for (const Refr& r : refs) {
                int b0 = std::min(NB - 1, std::max(0, (int)((r.box.lo[axis] - lo) * inv)));
                int b1 = std::min(NB - 1, std::max(0, (int)((r.box.hi[axis] - lo) * inv)));
                ++entry[b0]; ++exit_[b1];
                bool sph = (ctx.prim_obj[r.prim] & PRIM_SPHERE_BIT) != 0;
                for (int b = b0; b <= b1; ++b) {
                    AABB cb = (b0 == b1) ? r.box
                        : clip_tri_to_slab(ctx.prims[r.prim], sph, r.box, axis,
                                           lo + b * width, lo + (b + 1) * width);
                    if (cb.valid()) bins[b].grow(cb);
                }
            }
            AABB lb[NB]; int lc[NB];
            AABB acc; int c = 0;
            for (int b = 0; b < NB; ++b) { acc.grow(bins[b]); c += entry[b]; lb[b] = acc; lc[b] = c; }
            AABB rb[NB]; AABB racc;
            int rcnt[NB];
            int rc = 0;
            for (int b = NB - 1; b >= 1; --b) { racc.grow(bins[b]); rb[b] = racc; }
            rc = 0;
            for (int b = NB - 1; b >= 1; --b) { rc += exit_[b]; rcnt[b] = rc; }
            for (int b = 0; b < NB - 1; ++b) {
                int nl = lc[b], nr = rcnt[b + 1];
                if (nl == 0 || nr == 0) continue;
                float cost = lb[b].area() * nl + rb[b + 1].area() * nr;
                if (cost < best_sp_cost) { best_sp_cost = cost; best_sp_axis = axis; best_sp_bin = b; }
            }
        }
    }

    bool use_spatial = best_sp_axis >= 0 && best_sp_cost < best_obj_cost;
    if (!use_spatial && best_obj_axis < 0) {
        if (n <= 15) { make_leaf(); return; }  // leaf cap: ww walk packs cnt in 4 bits
        median_split();
        return;
    }
    if (n <= ctx.max_leaf &&
        std::min(best_obj_cost, best_sp_cost) + ctx.trav_cost * box.area() >= leaf_cost) {
        make_leaf();
        return;
    }

    std::vector<Refr> lrefs, rrefs;
    if (use_spatial) {
        ++out.n_spatial;
        int axis = best_sp_axis;
        float lo = box.lo[axis];
        float width = ext[axis] / NB;
        float plane = lo + (best_sp_bin + 1) * width;
        AABB lbox_all, rbox_all;
        // first pass boxes for unsplit cost estimation
        for (const Refr& r : refs) {
            if (r.box.hi[axis] <= plane) lbox_all.grow(r.box);
            else if (r.box.lo[axis] >= plane) rbox_all.grow(r.box);
        }
        for (const Refr& r : refs) {
            if (r.box.hi[axis] <= plane) { lrefs.push_back(r); continue; }
            if (r.box.lo[axis] >= plane) { rrefs.push_back(r); continue; }
            bool sph = (ctx.prim_obj[r.prim] & PRIM_SPHERE_BIT) != 0;
            AABB lb = clip_tri_to_slab(ctx.prims[r.prim], sph, r.box, axis, lo - 1.f, plane);
            AABB rb = clip_tri_to_slab(ctx.prims[r.prim], sph, r.box, axis, plane,
                                       box.hi[axis] + 1.f);
            if (ctx.ref_unsplit) {
                // reference unsplitting (bvh_spatial.cu:468-529): compare the
                // cost of duplicating vs pushing the whole ref to one side
                AABB lgrow = lbox_all; lgrow.grow(r.box);
                AABB rgrow = rbox_all; rgrow.grow(r.box);
                float c_split = lbox_all.area() + rbox_all.area();  // marginal proxies
                float c_left = lgrow.area() + rbox_all.area();
                float c_right = lbox_all.area() + rgrow.area();
                if (c_left <= c_split && c_left <= c_right) {
                    lrefs.push_back(r); lbox_all = lgrow; continue;
                }
                if (c_right < c_split && c_right < c_left) {
                    rrefs.push_back(r); rbox_all = rgrow; continue;
                }
            }
            if (lb.valid()) lrefs.push_back({lb, r.prim});
            if (rb.valid()) rrefs.push_back({rb, r.prim});
        }
    } else {
        int axis = best_obj_axis;
        float cmin = cbox.lo[axis];
        float inv = NB / cext[axis];
        for (const Refr& r : refs) {
            int b = std::min(NB - 1, (int)((r.box.centroid()[axis] - cmin) * inv));
            (b <= best_obj_bin ? lrefs : rrefs).push_back(r);
        }
    }
    if (lrefs.empty() || rrefs.empty()) {
        if (n <= 15) make_leaf(); else median_split();
        return;
    }
    refs.clear(); refs.shrink_to_fit();

    int li, ri;
    if (depth < 3 && (int)(lrefs.size() + rrefs.size()) > 16384) {
        BuildOut lout, rout;
        std::thread tl([&] { build_rec(ctx, std::move(lrefs), depth + 1, lout); });
        build_rec(ctx, std::move(rrefs), depth + 1, rout);
        tl.join();
        li = merge_child(out, lout);
        ri = merge_child(out, rout);
    } else {
        li = (int)out.nodes.size();
        build_rec(ctx, std::move(lrefs), depth + 1, out);
        ri = (int)out.nodes.size();
        build_rec(ctx, std::move(rrefs), depth + 1, out);
    }
    out.nodes[my].left = li;
    out.nodes[my].right = ri;
}

void linearize_s(const std::vector<SNode>& pool, const std::vector<Refr>& leaf_refs,
                 int root, std::vector<BVHNode>& out, std::vector<int>& order,
                 float* sah, float inv_root_area) {
    const SNode& nd = pool[root];
    int my = (int)out.size();
    out.emplace_back();
    out[my].lo = Vec4(nd.box.lo, 0.f);
    out[my].hi = Vec4(nd.box.hi, 0.f);
    if (nd.left < 0) {
        out[my].lo.w = int_as_float((int)order.size());
        out[my].hi.w = int_as_float(nd.leaf_cnt);
        for (int k = 0; k < nd.leaf_cnt; ++k) order.push_back(leaf_refs[nd.leaf_base + k].prim);
        *sah += nd.box.area() * inv_root_area * nd.leaf_cnt;
    } else {
        linearize_s(pool, leaf_refs, nd.left, out, order, sah, inv_root_area);
        linearize_s(pool, leaf_refs, nd.right, out, order, sah, inv_root_area);
        out[my].lo.w = int_as_float(-1);
        out[my].hi.w = int_as_float(-(int)out.size());
        *sah += pool[root].box.area() * inv_root_area * 1.2f;
    }
}

} // namespace

BVHBuildResult build_sbvh(const Prim* prims, const uint32_t* prim_obj, int n,
                          const BVHBuildConfig& cfg) {
    BVHBuildResult res;
    if (n <= 0) return res;
    std::vector<Refr> refs(n);
    AABB root_box;
    for (int i = 0; i < n; ++i) {
        AABB box;
        if (prim_obj[i] & PRIM_SPHERE_BIT) {
            Vec3 c = prims[i].v0.xyz(); float r = prims[i].v0.w;
            box.grow(c - Vec3(r)); box.grow(c + Vec3(r));
        } else {
            Vec3 v0 = prims[i].v0.xyz();
            box.grow(v0);
            box.grow(v0 + prims[i].e1.xyz());
            box.grow(v0 + prims[i].e2.xyz());
        }
        refs[i] = {box, i};
        root_box.grow(box);
    }
    Ctx ctx{prims, prim_obj, std::max(1, std::min(cfg.max_leaf_prims, 15)), cfg.ref_unsplit,
            root_box.area() > 0.f ? 1.f / root_box.area() : 0.f, cfg.trav_cost};
    BuildOut out;
    out.nodes.reserve(2 * n);
    out.leaf_refs.reserve((size_t)(n * 1.3));
    build_rec(ctx, std::move(refs), 0, out);
    res.nodes.reserve(out.nodes.size());
    res.prim_order.reserve(out.leaf_refs.size());
    float sah = 0.f;
    linearize_s(out.nodes, out.leaf_refs, 0, res.nodes, res.prim_order, &sah,
                ctx.inv_root_area);
    res.sah_cost = sah;
    res.n_leaves = out.n_leaves;
    res.max_depth = out.max_depth;
    return res;
}

} // namespace hippt
