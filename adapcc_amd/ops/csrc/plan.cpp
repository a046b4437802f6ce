// Host-only unit-plan builders for every primitive (see plan.h).
//
// All plans are deterministic functions of (shape/world, sizes, active mask)
// so every rank independently derives the same flag grid. Single-node direct
// algorithms (every pair of MI355X GPUs has a dedicated xGMI link):
//   allreduce      tree forest: chunked reduce phase + broadcast phase
//   reduce         forest reduce phase + per-slice gather at the root
//   broadcast      receivers pull root's staged slices directly
//   allgather      receivers pull each source's staged tensor directly
//   reducescatter  star forest reduce phase, slice t owned by rank t
//   alltoall       (src,dst)-tiled direct pulls from the source's sendbuf
// Reference parity: these were declared but unimplemented upstream
// (trans.h:31-33); reduce/broadcast existed as tree contexts (reduce.cu,
// boardcast.cu).

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

TreeShape star_shape(int world) {
  std::vector<std::vector<int>> parents(world, std::vector<int>(world));
  for (int t = 0; t < world; ++t)
    for (int r = 0; r < world; ++r) parents[t][r] = (r == t) ? -1 : t;
  return TreeShape::derive(parents);
}

namespace {

constexpr long kAlignE = 64;

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

// Relay / effective-source analysis (reference: csrc/control.cu:27-101):
// inactive node + single active inflow -> skipped (consumer pulls the inflow
// directly over xGMI); >=2 inflows -> still aggregates; the root always
// materializes its tree's result.
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

std::vector<char> mask_to_active(uint64_t mask, int world) {
  std::vector<char> active(world, 0);
  for (int r = 0; r < world; ++r)
    if (mask & (1ull << r)) active[r] = 1;
  return active;
}

long pick_chunk_elems(long slice_elems, int esize, long chunk_bytes,
                      int max_slots = kMaxChunkSlots) {
  long ce = std::max<long>(chunk_bytes / esize, kAlignE);
  ce = (ce + kAlignE - 1) / kAlignE * kAlignE;
  while ((slice_elems + ce - 1) / ce > max_slots) ce *= 2;
  return ce;
}

struct TC {
  int t;
  int c;
  long off;
  long cnt;
};

// Per-tree slice ranges over [0, total): equal by default; with weights,
// boundaries at cumulative fractions (64-element aligned, monotone,
// degenerate/empty slices allowed). Deterministic across ranks.
std::vector<std::pair<long, long>> make_slices(
    int T, long total, const std::vector<double>& weights) {
  std::vector<std::pair<long, long>> slice(T);
  if (weights.empty()) {
    const long per_raw = (total + T - 1) / T;
    const long per = (per_raw + kAlignE - 1) / kAlignE * kAlignE;
    for (int t = 0; t < T; ++t) {
      slice[t] = {std::min((long)t * per, total),
                  std::min((long)(t + 1) * per, total)};
    }
    return slice;
  }
  if ((int)weights.size() != T)
    throw std::runtime_error("slice_weights length != num trees");
  double sum = 0.0;
  for (double w : weights) {
    if (!(w > 0.0)) throw std::runtime_error("slice_weights must be > 0");
    sum += w;
  }
  double acc = 0.0;
  long prev = 0;
  for (int t = 0; t < T; ++t) {
    acc += weights[t];
    long end = (t == T - 1)
                   ? total
                   : std::min<long>(
                         total,
                         (long)(acc / sum * total + 0.5) / kAlignE * kAlignE);
    if (end < prev) end = prev;
    slice[t] = {prev, end};
    prev = end;
  }
  return slice;
}

// (chunk, tree)-ordered grid over per-tree slices [beg, end).
std::vector<TC> make_grid(const std::vector<std::pair<long, long>>& slice,
                          long chunk_elems) {
  std::vector<TC> grid;
  long max_chunks = 0;
  for (const auto& s : slice) {
    long n = (s.second - s.first + chunk_elems - 1) / chunk_elems;
    max_chunks = std::max(max_chunks, n);
  }
  for (long c = 0; c < max_chunks; ++c) {
    for (int t = 0; t < (int)slice.size(); ++t) {
      long beg = slice[t].first + c * chunk_elems;
      if (beg >= slice[t].second) continue;
      grid.push_back({t, (int)c, beg, std::min(chunk_elems, slice[t].second - beg)});
    }
  }
  return grid;
}

void check_units(const PlanData& p) {
  const long maxu = (long)kMaxTrees * kMaxChunkSlots;
  if ((long)p.cunits.size() > maxu || (long)p.runits.size() > maxu ||
      (long)p.bunits.size() > maxu)
    throw std::runtime_error("too many units");
}

// Shared forest reduce phase (copy-in + reduce units); publishes each tree's
// materialized root result to `publish_to(tree_root)`.
void forest_reduce_phase(const TreeShape& shape, int rank,
                         const std::vector<TC>& grid,
                         const std::vector<TreeAnalysis>& ana,
                         const std::vector<char>& active,
                         const std::vector<std::vector<int>>& publish_to,
                         PlanData& plan) {
  const bool me_active = active[rank] != 0;
  for (const auto& tc : grid) {
    const TreeAnalysis& A = ana[tc.t];
    if (me_active) {
      CopyUnit cu{};
      cu.tree = tc.t;
      cu.chunk = tc.c;
      cu.offset_elems = tc.off;
      cu.count_elems = tc.cnt;
      cu.flag_space = 0;
      auto sc = A.send_consumer.find(rank);
      if (sc != A.send_consumer.end()) {
        cu.nnotify = 1;
        cu.notify_rank[0] = sc->second;
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
        for (int k : publish_to[tc.t]) ru.child_rank[ru.nchildren++] = k;
      }
      plan.runits.push_back(ru);
    }
  }
}

}  // namespace


// Partial active set: collapse each tree to a star rooted at an ACTIVE
// rank, so no excluded rank aggregates or forwards (an excluded straggler
// would otherwise sit on the actives' critical path; round-1 verdict
// item 8). Excluded ranks stay in the tree as leaves: they contribute
// nothing to the reduce (analyze_subtree skips them) but can still pull
// the result in the broadcast phase. Full-mask plans keep the strategy's
// shapes unchanged. `prefer_root`: re-root target (-1 = lowest active).
TreeShape effective_shape(const TreeShape& shape,
                          const std::vector<char>& active,
                          int prefer_root = -1) {
  const int world = shape.world;
  bool full = true;
  for (int r = 0; r < world; ++r)
    if (!active[r]) { full = false; break; }
  if (full) return shape;
  int fallback = prefer_root;
  if (fallback < 0 || !active[fallback]) {
    fallback = -1;
    for (int r = 0; r < world; ++r)
      if (active[r]) { fallback = r; break; }
  }
  if (fallback < 0) return shape;  // no active rank: leave as-is
  const int T = (int)shape.parents.size();
  std::vector<std::vector<int>> par(T, std::vector<int>(world));
  for (int t = 0; t < T; ++t) {
    int root = shape.roots[t];
    if (!active[root]) root = fallback;
    for (int r = 0; r < world; ++r) par[t][r] = (r == root) ? -1 : root;
  }
  return TreeShape::derive(par);
}

PlanData build_plan(const TreeShape& shape, int rank, long total_elems,
                    int esize, long chunk_bytes, uint64_t active_mask,
                    const std::vector<double>& slice_weights) {
  const int world = shape.world;
  auto active = mask_to_active(active_mask, world);
  const TreeShape eff = effective_shape(shape, active);
  const int T = (int)eff.parents.size();

  auto slice = make_slices(T, total_elems, slice_weights);
  long max_slice = 0;
  for (const auto& s : slice) max_slice = std::max(max_slice, s.second - s.first);
  const long chunk_elems = pick_chunk_elems(max_slice, esize, chunk_bytes);

  PlanData plan;
  plan.chunk_elems = chunk_elems;

  std::vector<TreeAnalysis> ana(T);
  std::vector<std::vector<int>> publish(T);
  for (int t = 0; t < T; ++t) {
    ana[t].root = eff.roots[t];
    analyze_subtree(eff.roots[t], eff.children[t], active, ana[t]);
    // allreduce: root publishes to itself + its direct children
    publish[t].push_back(eff.roots[t]);
    for (int c : eff.children[t][eff.roots[t]]) publish[t].push_back(c);
  }

  auto grid = make_grid(slice, chunk_elems);
  forest_reduce_phase(eff, rank, grid, ana, active, publish, plan);

  // broadcast phase: every rank receives every chunk
  for (const auto& tc : grid) {
    const TreeAnalysis& A = ana[tc.t];
    BcastUnit bu{};
    bu.tree = tc.t;
    bu.chunk = tc.c;
    bu.src_offset_elems = tc.off;
    bu.dst_offset_elems = tc.off;
    bu.count_elems = tc.cnt;
    if (rank == A.root) {
      bu.parent_rank = -1;
      bu.parent_kind = (uint8_t)BufKind::Acc;
    } else {
      int p = eff.parents[tc.t][rank];
      bu.parent_rank = p;
      bu.parent_kind = (uint8_t)(p == A.root ? BufKind::Acc : BufKind::Result);
    }
    const auto& kids = eff.children[tc.t][rank];
    if (rank != A.root && !kids.empty()) {
      bu.forward = 1;
      for (int c : kids) bu.child_rank[bu.nchildren++] = c;
    }
    plan.bunits.push_back(bu);
  }
  check_units(plan);
  return plan;
}

PlanData build_reduce_plan(const TreeShape& shape, int rank, int root,
                           long total_elems, int esize, long chunk_bytes,
                           uint64_t active_mask,
                           const std::vector<double>& slice_weights) {
  const int world = shape.world;
  if (root < 0 || root >= world) throw std::runtime_error("reduce: bad root");
  auto active = mask_to_active(active_mask, world);
  const TreeShape eff = effective_shape(shape, active, root);
  const int T = (int)eff.parents.size();

  auto slice = make_slices(T, total_elems, slice_weights);
  long max_slice = 0;
  for (const auto& s : slice) max_slice = std::max(max_slice, s.second - s.first);
  const long chunk_elems = pick_chunk_elems(max_slice, esize, chunk_bytes);

  PlanData plan;
  plan.chunk_elems = chunk_elems;

  std::vector<TreeAnalysis> ana(T);
  std::vector<std::vector<int>> publish(T);
  for (int t = 0; t < T; ++t) {
    ana[t].root = eff.roots[t];
    analyze_subtree(eff.roots[t], eff.children[t], active, ana[t]);
    publish[t] = {root};  // each tree's result goes to THE root only
  }
  auto grid = make_grid(slice, chunk_elems);
  forest_reduce_phase(eff, rank, grid, ana, active, publish, plan);

  if (rank == root) {
    for (const auto& tc : grid) {
      BcastUnit bu{};
      bu.tree = tc.t;
      bu.chunk = tc.c;
      bu.src_offset_elems = tc.off;
      bu.dst_offset_elems = tc.off;
      bu.count_elems = tc.cnt;
      int troot = eff.roots[tc.t];
      bu.parent_rank = (troot == rank) ? -1 : troot;
      bu.parent_kind = (uint8_t)BufKind::Acc;
      plan.bunits.push_back(bu);
    }
  }
  check_units(plan);
  return plan;
}

PlanData build_broadcast_plan(int world, int rank, int root, long total_elems,
                              int esize, long chunk_bytes) {
  if (root < 0 || root >= world) throw std::runtime_error("broadcast: bad root");
  const int T = std::min<long>(std::min(world, 8), kMaxTrees);
  const long per_raw = (total_elems + T - 1) / T;
  const long per = (per_raw + kAlignE - 1) / kAlignE * kAlignE;
  const long chunk_elems = pick_chunk_elems(per, esize, chunk_bytes);

  PlanData plan;
  plan.chunk_elems = chunk_elems;
  std::vector<std::pair<long, long>> slice(T);
  for (int t = 0; t < T; ++t) {
    slice[t] = {std::min((long)t * per, total_elems),
                std::min((long)(t + 1) * per, total_elems)};
  }
  auto grid = make_grid(slice, chunk_elems);
  for (const auto& tc : grid) {
    if (rank == root) {
      CopyUnit cu{};
      cu.tree = tc.t;
      cu.chunk = tc.c;
      cu.offset_elems = tc.off;
      cu.count_elems = tc.cnt;
      cu.flag_space = 1;  // receivers pull directly
      for (int r = 0; r < world; ++r)
        if (r != root) cu.notify_rank[cu.nnotify++] = r;
      plan.cunits.push_back(cu);
    } else {
      BcastUnit bu{};
      bu.tree = tc.t;
      bu.chunk = tc.c;
      bu.src_offset_elems = tc.off;
      bu.dst_offset_elems = tc.off;
      bu.count_elems = tc.cnt;
      bu.parent_rank = root;
      bu.parent_kind = (uint8_t)BufKind::Send;
      plan.bunits.push_back(bu);
    }
  }
  check_units(plan);
  return plan;
}

PlanData build_allgather_plan(int world, int rank, long in_elems, int esize,
                              long chunk_bytes) {
  // tree s = source rank s; each source stages its in tensor; everyone
  // pulls each source's chunks into out[s*L ..]
  const long L = in_elems;
  const long chunk_elems = pick_chunk_elems(L, esize, chunk_bytes);
  PlanData plan;
  plan.chunk_elems = chunk_elems;
  const long nch = (L + chunk_elems - 1) / chunk_elems;
  for (long c = 0; c < nch; ++c) {
    const long off = c * chunk_elems;
    const long cnt = std::min(chunk_elems, L - off);
    for (int s = 0; s < world; ++s) {
      if (s == rank) {
        CopyUnit cu{};
        cu.tree = s;
        cu.chunk = (int)c;
        cu.offset_elems = off;
        cu.count_elems = cnt;
        cu.flag_space = 1;
        for (int r = 0; r < world; ++r) cu.notify_rank[cu.nnotify++] = r;
        plan.cunits.push_back(cu);
      }
      BcastUnit bu{};
      bu.tree = s;
      bu.chunk = (int)c;
      bu.src_offset_elems = off;
      bu.dst_offset_elems = (long)s * L + off;
      bu.count_elems = cnt;
      bu.parent_rank = (s == rank) ? -1 : s;
      bu.parent_kind = (uint8_t)BufKind::Send;
      plan.bunits.push_back(bu);
    }
  }
  check_units(plan);
  return plan;
}

PlanData build_reducescatter_plan(int world, int rank, long out_elems,
                                  int esize, long chunk_bytes,
                                  uint64_t active_mask) {
  // star forest with slice t = rank t's out range [t*L, (t+1)*L) of the
  // world*L input
  const long L = out_elems;
  auto shape = star_shape(world);
  auto active = mask_to_active(active_mask, world);
  const long chunk_elems = pick_chunk_elems(L, esize, chunk_bytes);

  PlanData plan;
  plan.chunk_elems = chunk_elems;
  std::vector<TreeAnalysis> ana(world);
  std::vector<std::vector<int>> publish(world);
  for (int t = 0; t < world; ++t) {
    ana[t].root = t;
    analyze_subtree(t, shape.children[t], active, ana[t]);
    publish[t] = {t};
  }
  std::vector<std::pair<long, long>> slice(world);
  for (int t = 0; t < world; ++t) slice[t] = {(long)t * L, (long)(t + 1) * L};
  auto grid = make_grid(slice, chunk_elems);
  forest_reduce_phase(shape, rank, grid, ana, active, publish, plan);

  for (const auto& tc : grid) {
    if (tc.t != rank) continue;
    BcastUnit bu{};
    bu.tree = tc.t;
    bu.chunk = tc.c;
    bu.src_offset_elems = tc.off;
    bu.dst_offset_elems = tc.off - (long)rank * L;
    bu.count_elems = tc.cnt;
    bu.parent_rank = -1;
    bu.parent_kind = (uint8_t)BufKind::Acc;
    plan.bunits.push_back(bu);
  }
  check_units(plan);
  return plan;
}

PlanData build_alltoall_plan(int world, int rank, long per_rank_elems,
                             int esize, long chunk_bytes) {
  // (src s, dst d) tiles: s stages its whole in; d pulls in[d*L..] of s into
  // out[s*L..]. Flag slot = d*C + c (C chunks per L-slice), tree = s.
  const long L = per_rank_elems;
  const long chunk_elems =
      pick_chunk_elems(L, esize, chunk_bytes, kMaxChunkSlots / world);
  const long C = (L + chunk_elems - 1) / chunk_elems;
  PlanData plan;
  plan.chunk_elems = chunk_elems;
  for (long c = 0; c < C; ++c) {
    const long loff = c * chunk_elems;
    const long cnt = std::min(chunk_elems, L - loff);
    for (int d = 0; d < world; ++d) {
      const int slot = (int)(d * C + c);
      {  // my copy unit for destination d
        CopyUnit cu{};
        cu.tree = rank;
        cu.chunk = slot;
        cu.offset_elems = (long)d * L + loff;
        cu.count_elems = cnt;
        cu.flag_space = 1;
        cu.nnotify = 1;
        cu.notify_rank[0] = d;
        plan.cunits.push_back(cu);
      }
      if (d == rank) {
        // I receive from every source s the range [rank*L ..] of s's in
        for (int s = 0; s < world; ++s) {
          BcastUnit bu{};
          bu.tree = s;
          bu.chunk = slot;
          bu.src_offset_elems = (long)rank * L + loff;
          bu.dst_offset_elems = (long)s * L + loff;
          bu.count_elems = cnt;
          bu.parent_rank = (s == rank) ? -1 : s;
          bu.parent_kind = (uint8_t)BufKind::Send;
          plan.bunits.push_back(bu);
        }
      }
    }
  }
  check_units(plan);
  return plan;
}

}  // namespace adapcc
