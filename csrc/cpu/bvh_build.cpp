// bvh_build.cpp — binned SAH BVH builder with overlap penalty, multithreaded
// top-level splits, and skip-link linearization.
//
// Capability parity: reference src/impl/bvh.cu (binned SAH :170-186, fallback
// equal-count split, leaf threshold, primitive reordering :329-401).  The
// linearized output uses an explicit DFS skip-link (the reference encodes the
// same traversal as negative subtree offsets, bvh.cuh:341-377).
#include "bvh_build.h"
#include <algorithm>
#include <mutex>
#include <atomic>
#include <thread>
#include <cmath>

namespace hippt {

namespace {

constexpr int N_BINS = 16;

struct BuildPrim {
    AABB box;
    Vec3 centroid;
    int idx;
};

struct BuildNode {
    AABB box;
    int left = -1, right = -1;  // children indices into node pool
    int prim_base = 0, prim_cnt = 0;
    bool leaf() const { return left < 0; }
};

struct Builder {
    std::vector<BuildPrim> bp;
    std::vector<BuildNode> pool;
    std::mutex pool_mu;
    int max_leaf;
    float overlap_w;
    float trav_cost;

    int alloc_node() {
        std::lock_guard<std::mutex> g(pool_mu);
        pool.emplace_back();
        return (int)pool.size() - 1;
    }

    // Binned SAH split of bp[lo,hi); returns partition point or -1 for leaf.
    int find_split(int lo, int hi, const AABB& box, int& axis_out) {
        int n = hi - lo;
        AABB cbox;
        for (int i = lo; i < hi; ++i) cbox.grow(bp[i].centroid);
        Vec3 ext = cbox.extent();
        int axis = 0;
        if (ext.y > ext.x) axis = 1;
        if (ext.z > ext[axis]) axis = 2;
        axis_out = axis;
        float cmin = cbox.lo[axis], cext = ext[axis];
        if (cext < 1e-12f) return -1;  // all centroids coincide -> leaf/equal split

        AABB bins[N_BINS];
        int cnt[N_BINS] = {0};
        float inv = N_BINS / cext;
        for (int i = lo; i < hi; ++i) {
            int b = std::min(N_BINS - 1, (int)((bp[i].centroid[axis] - cmin) * inv));
            bins[b].grow(bp[i].box);
            ++cnt[b];
        }
        // sweep
        AABB lbox[N_BINS]; int lcnt[N_BINS];
        AABB acc; int c = 0;
        for (int b = 0; b < N_BINS; ++b) { acc.grow(bins[b]); c += cnt[b]; lbox[b] = acc; lcnt[b] = c; }
        AABB racc; float best = 1e30f; int best_b = -1;
        AABB rbox_best;
        int rc = 0;
        AABB rboxes[N_BINS];
        for (int b = N_BINS - 1; b >= 1; --b) { racc.grow(bins[b]); rboxes[b] = racc; }
        for (int b = 0; b < N_BINS - 1; ++b) {
            int nl = lcnt[b], nr = n - nl;
            if (nl == 0 || nr == 0) continue;
            float cost = lbox[b].area() * nl + rboxes[b + 1].area() * nr;
            // overlap penalty (reference bvh.cu:170-186)
            if (overlap_w > 0.5f)
                cost += AABB::intersection_area(lbox[b], rboxes[b + 1]) * n * (overlap_w - 0.5f);
            if (cost < best) { best = cost; best_b = b; rbox_best = rboxes[b + 1]; }
        }
        float leaf_cost = box.area() * n;
        if (best_b < 0 || (n <= max_leaf && best + trav_cost * box.area() >= leaf_cost))
            return -1;
        // partition
        auto mid = std::partition(bp.begin() + lo, bp.begin() + hi, [&](const BuildPrim& p) {
            int b = std::min(N_BINS - 1, (int)((p.centroid[axis] - cmin) * inv));
            return b <= best_b;
        });
        int m = (int)(mid - bp.begin());
        if (m == lo || m == hi) return -1;
        return m;
    }

    int build_range(int lo, int hi, int depth, int* out_depth) {
        int ni = alloc_node();
        AABB box;
        for (int i = lo; i < hi; ++i) box.grow(bp[i].box);
        int n = hi - lo;
        *out_depth = std::max(*out_depth, depth);
        if (n <= max_leaf) {
            int axis;
            int m = n > 1 ? find_split(lo, hi, box, axis) : -1;
            if (m < 0) {
                std::lock_guard<std::mutex> g(pool_mu);
                pool[ni].box = box; pool[ni].prim_base = lo; pool[ni].prim_cnt = n;
                return ni;
            }
            int l = build_range(lo, m, depth + 1, out_depth);
            int r = build_range(m, hi, depth + 1, out_depth);
            std::lock_guard<std::mutex> g(pool_mu);
            pool[ni].box = box; pool[ni].left = l; pool[ni].right = r;
            return ni;
        }
        int axis;
        int m = find_split(lo, hi, box, axis);
        if (m < 0) {
            if (n > 15) {
                // forced equal-count split: leaves are capped at 15 prims so
                // the while-while walk can pack (cnt,base) into 31 bits
                m = lo + n / 2;
                std::nth_element(bp.begin() + lo, bp.begin() + m, bp.begin() + hi,
                                 [&](const BuildPrim& a, const BuildPrim& b_) {
                                     return a.centroid[axis] < b_.centroid[axis];
                                 });
            } else {
                std::lock_guard<std::mutex> g(pool_mu);
                pool[ni].box = box; pool[ni].prim_base = lo; pool[ni].prim_cnt = n;
                return ni;
            }
        }
        int l, r;
        if (depth < 3 && n > 16384) {
            // parallel top-level splits (reference SBVH thread pool analog)
            int dl = 0, dr = 0;
            std::thread tl([&] { l = build_range(lo, m, depth + 1, &dl); });
            r = build_range(m, hi, depth + 1, &dr);
            tl.join();
            *out_depth = std::max(*out_depth, std::max(dl, dr));
        } else {
            l = build_range(lo, m, depth + 1, out_depth);
            r = build_range(m, hi, depth + 1, out_depth);
        }
        std::lock_guard<std::mutex> g(pool_mu);
        pool[ni].box = box; pool[ni].left = l; pool[ni].right = r;
        return ni;
    }
};

// DFS linearization with skip links.
void linearize(const std::vector<BuildNode>& pool, int root, std::vector<BVHNode>& out,
               int* n_leaves, float* sah, float root_area) {
    struct Item { int node; };
    // recursive lambda via explicit stack of (node, phase)
    std::vector<std::pair<int, int>> stack;  // (pool idx, out idx placeholder)
    // simple recursion
    struct Rec {
        const std::vector<BuildNode>& pool;
        std::vector<BVHNode>& out;
        int* n_leaves; float* sah; float inv_root_area;
        void go(int ni) {
            const BuildNode& nd = pool[ni];
            int my = (int)out.size();
            out.emplace_back();
            BVHNode& ln = out[my];
            ln.lo = Vec4(nd.box.lo, 0.f);
            ln.hi = Vec4(nd.box.hi, 0.f);
            if (nd.leaf()) {
                ln.lo.w = int_as_float(nd.prim_base);
                ln.hi.w = int_as_float(nd.prim_cnt);
                ++*n_leaves;
                *sah += nd.box.area() * inv_root_area * nd.prim_cnt;
            } else {
                go(nd.left);
                go(nd.right);
                out[my].lo.w = int_as_float(-1);
                out[my].hi.w = int_as_float(-(int)out.size());  // skip = after subtree
                *sah += pool[ni].box.area() * inv_root_area * 1.2f;
            }
        }
    } rec{pool, out, n_leaves, sah, root_area > 0.f ? 1.f / root_area : 0.f};
    rec.go(root);
}

} // namespace

BVHBuildResult build_bvh(const Prim* prims, const uint32_t* prim_obj, int n,
                         const BVHBuildConfig& cfg) {
    BVHBuildResult res;
    if (n <= 0) return res;
    Builder b;
    b.max_leaf = std::max(1, std::min(cfg.max_leaf_prims, 15));
    b.overlap_w = cfg.overlap_w;
    b.trav_cost = cfg.trav_cost;
    b.bp.resize(n);
    for (int i = 0; i < n; ++i) {
        AABB box;
        if (prim_obj[i] & PRIM_SPHERE_BIT) {
            Vec3 c = prims[i].v0.xyz();
            float r = prims[i].v0.w;
            box.grow(c - Vec3(r)); box.grow(c + Vec3(r));
        } else {
            Vec3 v0 = prims[i].v0.xyz();
            box.grow(v0);
            box.grow(v0 + prims[i].e1.xyz());
            box.grow(v0 + prims[i].e2.xyz());
        }
        b.bp[i] = {box, box.centroid(), i};
    }
    b.pool.reserve(2 * n);
    int depth = 0;
    int root = b.build_range(0, n, 0, &depth);
    res.max_depth = depth;
    res.nodes.reserve(2 * n);
    float root_area = b.pool[root].box.area();
    linearize(b.pool, root, res.nodes, &res.n_leaves, &res.sah_cost, root_area);
    res.prim_order.resize(n);
    for (int i = 0; i < n; ++i) res.prim_order[i] = b.bp[i].idx;
    return res;
}

} // namespace hippt

// ---------------------------------------------------------------- BVH4 ----
namespace hippt {
namespace {

// Recover the two children of internal skip-link node i:
// left = i+1; right = skip(left) for internal left, left+1 for leaf left.
inline int right_child(const std::vector<BVHNode>& bin, int left) {
    return bin[left].is_leaf() ? left + 1 : bin[left].skip();
}

struct Collapser {
    const std::vector<BVHNode>& bin;
    std::vector<BVH4Node> out;
    int max_depth = 0;

    float area(int i) const {
        Vec3 e = bin[i].hi.xyz() - bin[i].lo.xyz();
        return 2.f * (e.x * e.y + e.y * e.z + e.z * e.x);
    }

    // emit the BVH4 node for binary node bi (must be internal unless root leaf)
    int emit(int bi, int depth) {
        max_depth = std::max(max_depth, depth);
        int my = (int)out.size();
        out.emplace_back();
        int slots[4];
        int n_slots = 0;
        if (bin[bi].is_leaf()) {
            slots[n_slots++] = bi;           // degenerate: whole tree is 1 leaf
        } else {
            int l = bi + 1, r = right_child(bin, l);
            slots[n_slots++] = l;
            slots[n_slots++] = r;
            // expand the largest-area internal slot until 4 slots
            while (n_slots < 4) {
                int pick = -1;
                float best = -1.f;
                for (int s = 0; s < n_slots; ++s)
                    if (!bin[slots[s]].is_leaf() && area(slots[s]) > best) {
                        best = area(slots[s]);
                        pick = s;
                    }
                if (pick < 0) break;
                int p = slots[pick];
                int pl = p + 1, pr = right_child(bin, pl);
                slots[pick] = pl;
                slots[n_slots++] = pr;
            }
        }
        // fill node (children recursively, in slot order = DFS)
        BVH4Node tmp{};
        for (int c = 0; c < 4; ++c) {
            if (c < n_slots) {
                int s = slots[c];
                tmp.lo_x[c] = bin[s].lo.x; tmp.lo_y[c] = bin[s].lo.y; tmp.lo_z[c] = bin[s].lo.z;
                tmp.hi_x[c] = bin[s].hi.x; tmp.hi_y[c] = bin[s].hi.y; tmp.hi_z[c] = bin[s].hi.z;
                if (bin[s].is_leaf()) {
                    tmp.child[c] = ~bin[s].prim_base();
                    tmp.cnt[c] = bin[s].prim_cnt();
                } else {
                    tmp.child[c] = emit(s, depth + 1);
                    tmp.cnt[c] = 0;
                }
            } else {
                // empty slot: degenerate far-away point box (a min/max slab
                // test un-inverts an inverted box, so inversion can't be the
                // never-hit encoding) + 0-prim leaf so a hit is still a no-op
                tmp.lo_x[c] = tmp.lo_y[c] = tmp.lo_z[c] = 3.0e38f;
                tmp.hi_x[c] = tmp.hi_y[c] = tmp.hi_z[c] = 3.0e38f;
                tmp.child[c] = ~0;
                tmp.cnt[c] = 0;
            }
        }
        out[my] = tmp;
        return my;
    }
};

} // namespace

std::vector<BVH4Node> collapse_bvh4(const std::vector<BVHNode>& bin, int* max_depth4) {
    Collapser col{bin};
    if (bin.empty()) return {};
    col.out.reserve(bin.size() / 2 + 1);
    col.emit(0, 1);
    if (max_depth4) *max_depth4 = col.max_depth;
    return col.out;
}

namespace {

// 8-wide collapse (same expansion rule, width 8; bvh8.h traversal).
struct Collapser8 {
    const std::vector<BVHNode>& bin;
    std::vector<BVH8Node> out;
    int max_depth = 0;

    float area(int i) const {
        Vec3 e = bin[i].hi.xyz() - bin[i].lo.xyz();
        return 2.f * (e.x * e.y + e.y * e.z + e.z * e.x);
    }

    int emit(int bi, int depth) {
        max_depth = std::max(max_depth, depth);
        int my = (int)out.size();
        out.emplace_back();
        int slots[8];
        int n_slots = 0;
        if (bin[bi].is_leaf()) {
            slots[n_slots++] = bi;
        } else {
            int l = bi + 1, r = right_child(bin, l);
            slots[n_slots++] = l;
            slots[n_slots++] = r;
            while (n_slots < 8) {
                int pick = -1;
                float best = -1.f;
                for (int s = 0; s < n_slots; ++s)
                    if (!bin[slots[s]].is_leaf() && area(slots[s]) > best) {
                        best = area(slots[s]);
                        pick = s;
                    }
                if (pick < 0) break;
                int p = slots[pick];
                int pl = p + 1, pr = right_child(bin, pl);
                slots[pick] = pl;
                slots[n_slots++] = pr;
            }
        }
        BVH8Node tmp{};
        for (int c = 0; c < 8; ++c) {
            if (c < n_slots) {
                int s = slots[c];
                tmp.lo_x[c] = bin[s].lo.x; tmp.lo_y[c] = bin[s].lo.y; tmp.lo_z[c] = bin[s].lo.z;
                tmp.hi_x[c] = bin[s].hi.x; tmp.hi_y[c] = bin[s].hi.y; tmp.hi_z[c] = bin[s].hi.z;
                if (bin[s].is_leaf()) {
                    tmp.child[c] = ~bin[s].prim_base();
                    tmp.cnt[c] = bin[s].prim_cnt();
                } else {
                    tmp.child[c] = emit(s, depth + 1);
                    tmp.cnt[c] = 0;
                }
            } else {
                tmp.lo_x[c] = tmp.lo_y[c] = tmp.lo_z[c] = 3.0e38f;
                tmp.hi_x[c] = tmp.hi_y[c] = tmp.hi_z[c] = 3.0e38f;
                tmp.child[c] = ~0;
                tmp.cnt[c] = 0;
            }
        }
        out[my] = tmp;
        return my;
    }
};

} // namespace

std::vector<BVH8Node> collapse_bvh8(const std::vector<BVHNode>& bin, int* max_depth8) {
    Collapser8 col{bin};
    if (bin.empty()) return {};
    col.out.reserve(bin.size() / 4 + 1);
    col.emit(0, 1);
    if (max_depth8) *max_depth8 = col.max_depth;
    return col.out;
}

} // namespace hippt
