// See plan.h. Extracted from the engine so CPU tests can drive it.

#include "plan.h"

#include <algorithm>
#include <map>
#include <stdexcept>

namespace adapcc {

TreeShape TreeShape::derive(const std::vector<std::vector<int>>& parents) {
  if (parents.empty()) throw std::runtime_error("empty strategy");
  if ((int)parents.size() > kMaxTrees) throw std::runtime_error("too many trees");
  TreeShape s;
  s.parents = parents;
  s.world = (int)parents[0].size();
  const int T = (int)parents.size();
  s.children.assign(T, std::vector<std::vector<int>>(s.world));
  s.roots.assign(T, -1);
  for (int t = 0; t < T; ++t) {
    if ((int)parents[t].size() != s.world)
      throw std::runtime_error("ragged parent arrays");
    for (int r = 0; r < s.world; ++r) {
      int p = parents[t][r];
      if (p < 0) {
        if (s.roots[t] >= 0) throw std::runtime_error("tree has two roots");
        s.roots[t] = r;
      } else {
        if (p >= s.world) throw std::runtime_error("parent out of range");
        s.children[t][p].push_back(r);
      }
    }
    if (s.roots[t] < 0) throw std::runtime_error("tree has no root");
    // cycle check: every rank must reach the root
    for (int r = 0; r < s.world; ++r) {
      int cur = r, hops = 0;
      while (cur != s.roots[t]) {
        cur = parents[t][cur];
        if (cur < 0 || ++hops > s.world)
          throw std::runtime_error("tree is not connected/acyclic");
      }
    }
  }
  return s;
}

namespace {

struct Provider {
  int rank;
  BufKind kind;
};

struct TreeAnalysis {
  struct Reducer {
    std::vector<Provider> srcs;
    bool include_self = false;
    int consumer = -1;
  };
  std::map<int, Reducer> reducers;
  std::map<int, int> send_consumer;
  int root = -1;
};

std::vector<Provider> analyze_subtree(int node,
                                      const std::vector<std::vector<int>>& children,
                                      const std::vector<char>& active,
                                      TreeAnalysis& out) {
  std::vector<Provider> below;
  for (int c : children[node]) {
    auto sub = analyze_subtree(c, children, active, out);
    below.insert(below.end(), sub.begin(), sub.end());
  }
  const bool self_active = active[node] != 0;
  const bool is_root = (node == out.root);
  const bool reduces =
      below.size() >= 2 || (self_active && !below.empty()) || is_root;
  if (reduces) {
    TreeAnalysis::Reducer red;
    red.srcs = below;
    red.include_self = self_active;
    for (const auto& p : below) {
      if (p.kind == BufKind::Send) {
        out.send_consumer[p.rank] = node;
      } else {
        out.reducers[p.rank].consumer = node;
      }
    }
    out.reducers[node] = std::move(red);
    return {{node, BufKind::Acc}};
  }
  if (self_active) return {{node, BufKind::Send}};
  return below;
}

}  // namespace

PlanData build_plan(const TreeShape& shape, int rank, long total_elems,
                    int esize, long chunk_bytes, uint64_t active_mask) {
  const int world = shape.world;
  const int T = (int)shape.parents.size();
  std::vector<char> active(world, 0);
  for (int r = 0; r < world; ++r)
    if (active_mask & (1ull << r)) active[r] = 1;

  const long align_e = 64;
  const long per_raw = (total_elems + T - 1) / T;
  const long per = ((per_raw + align_e - 1) / align_e) * align_e;
  long chunk_elems = std::max<long>(chunk_bytes / esize, align_e);
  chunk_elems = ((chunk_elems + align_e - 1) / align_e) * align_e;
  while ((per + chunk_elems - 1) / chunk_elems > kMaxChunkSlots) chunk_elems *= 2;

  PlanData plan;
  plan.chunk_elems = chunk_elems;

  std::vector<TreeAnalysis> ana(T);
  for (int t = 0; t < T; ++t) {
    ana[t].root = shape.roots[t];
    analyze_subtree(shape.roots[t], shape.children[t], active, ana[t]);
  }

  struct TC { int t; int c; long off; long cnt; };
  std::vector<TC> grid;
  long max_chunks = 0;
  std::vector<std::pair<long, long>> slice(T);
  for (int t = 0; t < T; ++t) {
    long beg = std::min((long)t * per, total_elems);
    long end = std::min(beg + per, total_elems);
    slice[t] = {beg, end};
    long n = (end - beg + chunk_elems - 1) / chunk_elems;
    max_chunks = std::max(max_chunks, n);
  }
  for (long c = 0; c < max_chunks; ++c) {
    for (int t = 0; t < T; ++t) {
      long beg = slice[t].first + c * chunk_elems;
      if (beg >= slice[t].second) continue;
      long cnt = std::min(chunk_elems, slice[t].second - beg);
      grid.push_back({t, (int)c, beg, cnt});
    }
  }

  const bool me_active = active[rank] != 0;
  for (const auto& tc : grid) {
    const TreeAnalysis& A = ana[tc.t];
    if (me_active) {
      CopyUnit cu{};
      cu.tree = tc.t;
      cu.chunk = tc.c;
      cu.offset_elems = tc.off;
      cu.count_elems = tc.cnt;
      auto sc = A.send_consumer.find(rank);
      if (sc != A.send_consumer.end()) {
        cu.notify_parent = 1;
        cu.parent_rank = sc->second;
      }
      plan.cunits.push_back(cu);
    }
    auto red = A.reducers.find(rank);
    if (red != A.reducers.end()) {
      ReduceUnit ru{};
      ru.tree = tc.t;
      ru.chunk = tc.c;
      ru.offset_elems = tc.off;
      ru.count_elems = tc.cnt;
      ru.nsrc = (int)red->second.srcs.size();
      if (ru.nsrc > kMaxSrcs) throw std::runtime_error("too many reduce sources");
      for (int s = 0; s < ru.nsrc; ++s) {
        ru.src_rank[s] = red->second.srcs[s].rank;
        ru.src_kind[s] = (uint8_t)red->second.srcs[s].kind;
      }
      ru.include_self = red->second.include_self ? 1 : 0;
      if (red->second.consumer >= 0) {
        ru.notify_parent = 1;
        ru.parent_rank = red->second.consumer;
      }
      if (rank == A.root) {
        ru.is_root = 1;
        ru.child_rank[ru.nchildren++] = rank;
        for (int c : shape.children[tc.t][rank])
          ru.child_rank[ru.nchildren++] = c;
      }
      plan.runits.push_back(ru);
    }
    {
      BcastUnit bu{};
      bu.tree = tc.t;
      bu.chunk = tc.c;
      bu.offset_elems = tc.off;
      bu.count_elems = tc.cnt;
      if (rank == A.root) {
        bu.parent_rank = -1;
        bu.parent_kind = (uint8_t)BufKind::Acc;
      } else {
        int p = shape.parents[tc.t][rank];
        bu.parent_rank = p;
        bu.parent_kind = (uint8_t)(p == A.root ? BufKind::Acc : BufKind::Result);
      }
      const auto& kids = shape.children[tc.t][rank];
      if (rank != A.root && !kids.empty()) {
        bu.forward = 1;
        for (int c : kids) bu.child_rank[bu.nchildren++] = c;
      }
      plan.bunits.push_back(bu);
    }
  }
  return plan;
}

}  // namespace adapcc
