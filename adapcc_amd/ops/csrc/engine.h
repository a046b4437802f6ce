// Engine class declaration (see engine.hip for design notes).
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <map>
#include <string>
#include <tuple>
#include <utility>
#include <vector>

#include "common.h"
#include "plan.h"

namespace adapcc {

class Engine {
 public:
  Engine(int rank, int world, int device, size_t cap_bytes, double timeout_ms);
  ~Engine();
  Engine(const Engine&) = delete;
  Engine& operator=(const Engine&) = delete;

  std::string ipc_handle() const;
  void connect(const std::vector<std::string>& handles,
               const std::vector<int>& peer_devices = {});
  // Test/emulation path: wire peers by raw device address (same process,
  // same device). Exercises the full flag protocol on one GPU.
  void connect_local(const std::vector<uintptr_t>& peer_addrs);
  uintptr_t region_addr() const { return (uintptr_t)region_; }
  void set_strategy(const std::vector<std::vector<int>>& parents,
                    long chunk_bytes,
                    const std::vector<double>& slice_weights = {});

  // Enqueue collectives on the caller stream (asynchronous; the caller
  // stream is made to depend on completion). active_ranks empty => all.
  void allreduce(void* data, long total_elems, int dtype, int op,
                 const std::vector<int>& active_ranks, bool average,
                 void* caller_stream);
  void reduce(void* data, long total_elems, int dtype, int op, int root,
              const std::vector<int>& active_ranks, void* caller_stream);
  void broadcast(void* data, long total_elems, int dtype, int root,
                 void* caller_stream);
  void all_gather(const void* in, void* out, long in_elems, int dtype,
                  void* caller_stream);
  void all_to_all(const void* in, void* out, long per_rank_elems, int dtype,
                  void* caller_stream);
  void reduce_scatter(const void* in, void* out, long out_elems, int dtype,
                      int op, const std::vector<int>& active_ranks,
                      bool average, void* caller_stream);

  void synchronize();
  std::pair<uint64_t, uint64_t> query_error();
  std::string dump_inbox();

  int rank() const { return rank_; }
  int world() const { return world_; }
  size_t capacity() const { return cap_bytes_; }
  int num_trees() const { return num_trees_; }

  struct Plan {
    std::vector<CopyUnit> cunits;
    std::vector<ReduceUnit> runits;
    std::vector<BcastUnit> bunits;
    CopyUnit* d_c = nullptr;
    ReduceUnit* d_r = nullptr;
    BcastUnit* d_b = nullptr;
    int* d_ranks = nullptr;
    int nranks = 0;
    long total_elems = 0;
    Dtype dt = Dtype::F32;
    void free_device();
  };

 private:
  struct PlanKey {
    int prim;
    long elems;
    int dt;
    int op;
    uint64_t mask;
    int root;
    bool operator<(const PlanKey& o) const {
      return std::tie(prim, elems, dt, op, mask, root) <
             std::tie(o.prim, o.elems, o.dt, o.op, o.mask, o.root);
    }
  };

  void build_tables();
  Plan& get_plan(int prim, long elems, Dtype dt, RedOp op, uint64_t active_mask,
                 int root);
  uint64_t resolve_mask(const std::vector<int>& active_ranks) const;
  void enqueue(const Plan& plan, const void* in, void* out, CallArgs& args,
               void* caller_stream);
  CallArgs make_args(Dtype dt, RedOp op, float scale, long elems);
  void check_ready(long bytes_needed) const;

  int rank_, world_, device_;
  size_t cap_bytes_;
  double timeout_ms_;
  uint64_t seq_ = 0;
  bool connected_ = false;
  bool local_peers_ = false;

  void* region_ = nullptr;
  size_t region_bytes_ = 0;
  size_t inbox_off_ = 0;
  size_t slot_bytes_ = 0;   // size of one call slot (pipelined mode: 2 slots)
  int n_slots_ = 2;         // cross-call overlap ON (A/B-measured); ADAPCC_PIPELINE=0 -> 1
  void* peer_base_[kMaxRanks];

  DevTables tabs_[2] = {};

  unsigned long long* counters_ = nullptr;
  uint64_t* h_err_ = nullptr;

  hipStream_t s_red_{}, s_bcast_{}, s_err_{};
  hipEvent_t ev_in_{}, ev_sync0_{}, ev_red_{}, ev_bc_{},
      ev_barrier_[2] = {};

  int num_trees_ = 0;
  long chunk_bytes_ = 4 * 1024 * 1024;
  TreeShape shape_;
  std::vector<double> slice_weights_;

  long small_fused_bytes_ = 256 * 1024;
  int wgs_per_group_ = 8;
  int n_groups_ = 16;

  std::map<PlanKey, Plan> plans_;
};

}  // namespace adapcc
