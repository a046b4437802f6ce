// Host-only unit-plan computation (no HIP dependency) so the exact logic
// the GPU engine runs can be unit-tested on CPU via the Python bindings.
//
// Relay-control semantics mirror the reference's controller truth tables
// (reference: csrc/control.cu:27-101): an inactive node with a single active
// inflow is skipped (its consumer pulls the inflow directly); with >=2
// inflows it still reduces (pure aggregation relay); an active node with
// inflows reduces and contributes.
#pragma once

#include <cstdint>
#include <vector>

#include "common.h"

namespace adapcc {

struct TreeShape {
  std::vector<std::vector<int>> parents;                 // [tree][rank]
  std::vector<std::vector<std::vector<int>>> children;   // [tree][rank][...]
  std::vector<int> roots;                                // [tree]
  int world = 0;
  // throws std::runtime_error on malformed input
  static TreeShape derive(const std::vector<std::vector<int>>& parents);
};

struct PlanData {
  std::vector<CopyUnit> cunits;
  std::vector<ReduceUnit> runits;
  std::vector<BcastUnit> bunits;
  long chunk_elems = 0;
};

// Deterministic for a given (shape, total_elems, esize, chunk_bytes, mask):
// every rank derives the same global (chunk, tree) grid, so flag pushes and
// waits agree across ranks by construction.
PlanData build_plan(const TreeShape& shape, int rank, long total_elems,
                    int esize, long chunk_bytes, uint64_t active_mask,
                    const std::vector<double>& slice_weights = {});

// Remaining primitives (single-node direct algorithms; see plan.cpp).
PlanData build_reduce_plan(const TreeShape& shape, int rank, int root,
                           long total_elems, int esize, long chunk_bytes,
                           uint64_t active_mask,
                           const std::vector<double>& slice_weights = {});
PlanData build_broadcast_plan(int world, int rank, int root, long total_elems,
                              int esize, long chunk_bytes);
PlanData build_allgather_plan(int world, int rank, long in_elems, int esize,
                              long chunk_bytes);
PlanData build_reducescatter_plan(int world, int rank, long out_elems,
                                  int esize, long chunk_bytes,
                                  uint64_t active_mask);
PlanData build_alltoall_plan(int world, int rank, long per_rank_elems,
                             int esize, long chunk_bytes);

// Fully-connected star forest (tree t rooted at rank t).
TreeShape star_shape(int world);

}  // namespace adapcc
