// bvh_build.h — host BVH construction API.
//
// Capability parity: reference src/impl/bvh.cu (16-bin centroid SAH with
// overlap penalty, equal-count fallback, leaf threshold, post-build primitive
// reordering) + src/core/bvh.cuh recursive_linearize.  SBVH (spatial splits)
// lives in sbvh_build.cpp with the same output contract.
#pragma once
#include <vector>
#include <cstdint>
#include "../core/bvh.h"
#include "../core/bvh4.h"
#include "../core/bvh8.h"

namespace hippt {

struct BVHBuildResult {
    std::vector<BVHNode> nodes;     // DFS-ordered skip-link nodes
    std::vector<int> prim_order;    // prim_order[new_slot] = original prim index
    int n_leaves = 0;
    int max_depth = 0;
    float sah_cost = 0.f;
};

struct BVHBuildConfig {
    int max_leaf_prims = 4;
    float overlap_w = 0.f;       // SAH overlap penalty weight (reference bvh_overlap_w)
    // SAH traversal-cost constant (units of one prim test): split only when
    // Ct*A_parent + sum A_i*N_i beats A_parent*N.  0 = reference behavior
    // (always split while area*count drops), which over-fragments leaves for
    // a 4-wide walk whose node step costs ~a dependent 128-B load.
    float trav_cost = 0.f;
    bool use_sbvh = false;       // spatial splits (SBVH)
    bool ref_unsplit = true;     // SBVH reference unsplitting
    int n_threads = 8;
};

// prims/n: primitive array (triangles use v0/e1/e2; spheres per prim_obj bit31)
BVHBuildResult build_bvh(const Prim* prims, const uint32_t* prim_obj, int n,
                         const BVHBuildConfig& cfg);

// Collapse a DFS skip-link binary BVH (build_bvh/build_sbvh output) into a
// 4-wide BVH (bvh4.h).  Children of each 4-wide node are found by repeatedly
// expanding the largest-area internal slot of the binary pair until 4 slots
// exist.  Returns nodes in DFS order (root = 0); max_depth4 reports the
// collapsed tree depth (traversal stack bound = 3 * depth).
std::vector<BVH4Node> collapse_bvh4(const std::vector<BVHNode>& bin,
                                    int* max_depth4 = nullptr);

// 8-wide collapse (256-byte nodes, bvh8.h).
std::vector<BVH8Node> collapse_bvh8(const std::vector<BVHNode>& bin,
                                    int* max_depth8 = nullptr);

// SBVH: spatial-split BVH (Stich et al. style chopped binning); may duplicate
// references, so prim_order can be longer than n.
BVHBuildResult build_sbvh(const Prim* prims, const uint32_t* prim_obj, int n,
                          const BVHBuildConfig& cfg);

} // namespace hippt
