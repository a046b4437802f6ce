// Standalone sanitizer harness for the unit planner (pure host C++):
// compiled with -fsanitize=address,undefined by tests/test_sanitize.py.
// The reference's data plane relied on benign-in-practice races
// (SURVEY §5); this build demonstrates the planner logic is ASan/UBSan
// clean across every primitive and a sweep of shapes/active sets.
#include <cassert>
#include <cstdio>
#include <vector>

#include "../../adapcc_amd/ops/csrc/plan.h"
#include "../../adapcc_amd/ops/csrc/plan.cpp"

using namespace adapcc;

static std::vector<std::vector<int>> chains(int n, int ntrees) {
  std::vector<std::vector<int>> out;
  for (int t = 0; t < ntrees; ++t) {
    std::vector<int> parents(n);
    std::vector<int> order(n);
    for (int i = 0; i < n; ++i) order[i] = (t + i) % n;
    parents[order[0]] = -1;
    for (int i = 1; i < n; ++i) parents[order[i]] = order[i - 1];
    out.push_back(parents);
  }
  return out;
}

int main() {
  for (int world : {2, 4, 8, 16}) {
    auto star = star_shape(world);
    auto chain = TreeShape::derive(chains(world, 2));
    for (long total : {1L, 64L, 1000L, 1L << 20}) {
      for (uint64_t mask : {(1ull << world) - 1, 1ull, (1ull << world) - 2}) {
        for (const auto& shape : {star, chain}) {
          auto p = build_plan(shape, world / 2, total, 4, 4096, mask);
          assert(!p.bunits.empty() || total == 0);
        }
        build_reduce_plan(star, 0, world - 1, total, 4, 4096, mask);
        build_reducescatter_plan(world, 1 % world, total, 4, 4096, mask);
      }
      build_broadcast_plan(world, 0, world - 1, total, 2, 4096);
      build_allgather_plan(world, 0, total, 2, 4096);
      build_alltoall_plan(world, world - 1, total, 4, 4096);
    }
  }
  // malformed inputs must throw, not corrupt
  int threw = 0;
  try {
    TreeShape::derive({{0, 1, 2, 3}});
  } catch (...) {
    ++threw;
  }
  try {
    build_reduce_plan(star_shape(4), 0, 9, 100, 4, 256, 0xf);
  } catch (...) {
    ++threw;
  }
  assert(threw == 2);
  std::printf("plan_sanitize ok\n");
  return 0;
}
