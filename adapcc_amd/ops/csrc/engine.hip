// adapcc_amd native engine — host side.
//
// MI355X-native replacement for the reference's communicator.so runtime
// (reference: csrc/run.cu, csrc/allreduce.cu, csrc/trans.cu, csrc/control.cu,
// csrc/shm_ipc.cpp). Key re-design decisions:
//  - single hipIpc-shared comm region per rank (send/acc/result + flag inbox)
//    instead of per-thread 1.6 GB staging buffers (ref init.h:19)
//  - handle exchange through the caller's bootstrap channel (Python
//    torch.distributed store) instead of POSIX/SysV shm + TCP port math
//  - the chunk pipeline runs ON the GPU (kernels.hip); the host computes a
//    per-(size, active-set) unit plan once, caches it, and enqueues 4 kernels
//  - relay control (ref control.cu truth tables) becomes effective-source
//    computation: inactive relays are *skipped* intra-node (the reducer pulls
//    the relay's sources directly over xGMI) rather than forwarded through
//  - calls are serialized by an on-device end-of-call barrier; every wait is
//    deadline-bounded so a wedged peer raises instead of hanging the GPU

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstring>
#include <map>
#include <mutex>
#include <cstdlib>
#include <stdexcept>
#include <string>
#include <tuple>
#include <vector>

#include "common.h"
#include "engine.h"
#include "plan.h"

namespace adapcc {

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error at " __FILE__ ":") +  \
                               std::to_string(__LINE__) + ": " +            \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)

// kernels.hip
void launch_collective(Dtype dt, const void* user, void* user_mut,
                       const CopyUnit* cunits, int nc, const ReduceUnit* runits,
                       int nr, const BcastUnit* bunits, int nb,
                       const DevTables& tabs, const CallArgs& args, int me,
                       unsigned long long* red_counters,
                       unsigned long long* bc_counters, int wgs_per_group,
                       int n_groups, hipStream_t s_red, hipStream_t s_bcast);
void launch_barrier(const DevTables& tabs, const CallArgs& args, int me, int world,
                    const int* ranks_dev, int nranks, hipStream_t stream);
void launch_small_fused(Dtype dt, const void* user, void* user_mut,
                        const CopyUnit* cunits, int nc,
                        const ReduceUnit* runits, int nr,
                        const BcastUnit* bunits, int nb, const DevTables& tabs,
                        const CallArgs& args, int me, int world,
                        const int* ranks_dev, int nranks, hipStream_t stream);

namespace {
constexpr int kMaxUnits = kMaxTrees * kMaxChunkSlots;
constexpr size_t kAlign = 256;
inline size_t align_up(size_t x, size_t a = kAlign) { return (x + a - 1) / a * a; }
}  // namespace

// ---------------------------------------------------------------------------
// Engine
// ---------------------------------------------------------------------------

Engine::Engine(int rank, int world, int device, size_t cap_bytes,
               double timeout_ms)
    : rank_(rank), world_(world), device_(device), cap_bytes_(align_up(cap_bytes)),
      timeout_ms_(timeout_ms) {
  if (world > kMaxRanks) throw std::runtime_error("world > kMaxRanks");
  HIP_CHECK(hipSetDevice(device_));
  // Cross-call pipelining default ON (slot-alternated buffers): measured
  // +6..15% allreduce busbw at 1-64 MB back-to-back calls (4-rank A/B,
  // profiles/README round 2). ADAPCC_PIPELINE=0 restores serial slots.
  if (const char* s = getenv("ADAPCC_PIPELINE")) n_slots_ = atoi(s) ? 2 : 1;
  // region: n_slots x ([send][acc][result][inbox]) — slot-alternated calls
  // overlap (call k's broadcast with call k+1's reduce) when pipelined
  inbox_off_ = 3 * cap_bytes_;
  slot_bytes_ = inbox_off_ + align_up(sizeof(FlagInbox));
  region_bytes_ = (size_t)n_slots_ * slot_bytes_;
  // dmabuf IPC handles for very large regions hang hipIpcOpenMemHandle on
  // this pool (observed at 3 GB); refuse loudly instead of wedging peers.
  if (region_bytes_ > (2ull << 30))
    throw std::runtime_error(
        "engine IPC region " + std::to_string(region_bytes_) +
        " B exceeds the 2 GiB dmabuf-IPC safety bound; lower ADAPCC_BUF_CAP"
        " or set ADAPCC_PIPELINE=0");
  HIP_CHECK(hipMalloc(&region_, region_bytes_));
  HIP_CHECK(hipMemset(region_, 0, region_bytes_));
  HIP_CHECK(hipMalloc(&counters_,
                      (size_t)n_slots_ * 3 * kMaxUnits *
                          sizeof(unsigned long long)));
  HIP_CHECK(hipMemset(counters_, 0,
                      (size_t)n_slots_ * 3 * kMaxUnits *
                          sizeof(unsigned long long)));
  HIP_CHECK(hipStreamCreateWithFlags(&s_red_, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&s_bcast_, hipStreamNonBlocking));
  HIP_CHECK(hipStreamCreateWithFlags(&s_err_, hipStreamNonBlocking));
  HIP_CHECK(hipEventCreateWithFlags(&ev_in_, hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&ev_sync0_, hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&ev_red_, hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&ev_bc_, hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&ev_barrier_[0], hipEventDisableTiming));
  HIP_CHECK(hipEventCreateWithFlags(&ev_barrier_[1], hipEventDisableTiming));
  HIP_CHECK(hipHostMalloc(&h_err_, 2 * sizeof(uint64_t)));
  for (int r = 0; r < kMaxRanks; ++r) peer_base_[r] = nullptr;
  peer_base_[rank_] = region_;
  if (const char* s = getenv("ADAPCC_SMALL_FUSED_BYTES"))
    small_fused_bytes_ = atol(s);
  if (const char* s = getenv("ADAPCC_WGS_PER_GROUP")) wgs_per_group_ = atoi(s);
  if (const char* s = getenv("ADAPCC_N_GROUPS")) n_groups_ = atoi(s);
  if (wgs_per_group_ < 1) wgs_per_group_ = 1;
  if (n_groups_ < 1) n_groups_ = 1;
}

Engine::~Engine() {
  if (!local_peers_) {
    for (int r = 0; r < world_; ++r) {
      if (r != rank_ && peer_base_[r]) (void)hipIpcCloseMemHandle(peer_base_[r]);
    }
  }
  if (region_) (void)hipFree(region_);
  if (counters_) (void)hipFree(counters_);
  if (h_err_) (void)hipHostFree(h_err_);
  for (auto& kv : plans_) kv.second.free_device();
  (void)hipStreamDestroy(s_red_);
  (void)hipStreamDestroy(s_bcast_);
  (void)hipStreamDestroy(s_err_);
  (void)hipEventDestroy(ev_in_);
  (void)hipEventDestroy(ev_sync0_);
  (void)hipEventDestroy(ev_red_);
  (void)hipEventDestroy(ev_bc_);
  (void)hipEventDestroy(ev_barrier_[0]);
  (void)hipEventDestroy(ev_barrier_[1]);
}

std::string Engine::ipc_handle() const {
  hipIpcMemHandle_t h;
  HIP_CHECK(hipIpcGetMemHandle(&h, region_));
  return std::string(reinterpret_cast<const char*>(&h), sizeof(h));
}

void Engine::connect(const std::vector<std::string>& handles,
                     const std::vector<int>& peer_devices) {
  if ((int)handles.size() != world_)
    throw std::runtime_error("connect: need one handle per rank");
  HIP_CHECK(hipSetDevice(device_));
  // Belt-and-braces for the multi-GPU case: enable peer access to every
  // distinct peer device before opening their IPC handles (the lazy flag
  // on hipIpcOpenMemHandle should do this on demand; being explicit makes
  // a misconfigured xGMI link fail loudly at setup, caught by self_test).
  for (int dev : peer_devices) {
    if (dev < 0 || dev == device_) continue;
    int can = 0;
    if (hipDeviceCanAccessPeer(&can, device_, dev) == hipSuccess && can) {
      hipError_t e = hipDeviceEnablePeerAccess(dev, 0);
      if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) {
        throw std::runtime_error(std::string("enable peer access to dev ") +
                                 std::to_string(dev) + ": " +
                                 hipGetErrorString(e));
      }
      (void)hipGetLastError();  // clear AlreadyEnabled sticky state
    }
  }
  for (int r = 0; r < world_; ++r) {
    if (r == rank_) continue;
    if (handles[r].size() != sizeof(hipIpcMemHandle_t))
      throw std::runtime_error("connect: bad handle size");
    hipIpcMemHandle_t h;
    std::memcpy(&h, handles[r].data(), sizeof(h));
    void* p = nullptr;
    HIP_CHECK(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess));
    peer_base_[r] = p;
  }
  connected_ = true;
  build_tables();
}

void Engine::connect_local(const std::vector<uintptr_t>& peer_addrs) {
  if ((int)peer_addrs.size() != world_)
    throw std::runtime_error("connect_local: need one address per rank");
  for (int r = 0; r < world_; ++r) {
    if (r == rank_) continue;
    peer_base_[r] = reinterpret_cast<void*>(peer_addrs[r]);
  }
  local_peers_ = true;
  connected_ = true;
  build_tables();
}

void Engine::build_tables() {
  for (int slot = 0; slot < n_slots_; ++slot) {
    for (int r = 0; r < world_; ++r) {
      char* base = static_cast<char*>(peer_base_[r]) + slot * slot_bytes_;
      tabs_[slot].send[r] = base;
      tabs_[slot].acc[r] = base + cap_bytes_;
      tabs_[slot].result[r] = base + 2 * cap_bytes_;
      tabs_[slot].inbox[r] = reinterpret_cast<FlagInbox*>(base + inbox_off_);
    }
    tabs_[slot].counters = counters_;
  }
}

void Engine::set_strategy(const std::vector<std::vector<int>>& parents,
                          long chunk_bytes,
                          const std::vector<double>& slice_weights) {
  shape_ = TreeShape::derive(parents);
  if (shape_.world != world_)
    throw std::runtime_error("parent array size != world");
  if (!slice_weights.empty() &&
      (int)slice_weights.size() != (int)parents.size())
    throw std::runtime_error("slice_weights length != num trees");
  num_trees_ = (int)shape_.parents.size();
  chunk_bytes_ = chunk_bytes;
  slice_weights_ = slice_weights;
  // strategy change invalidates cached plans
  for (auto& kv : plans_) kv.second.free_device();
  plans_.clear();
}

void Engine::Plan::free_device() {
  if (d_c) (void)hipFree(d_c);
  if (d_r) (void)hipFree(d_r);
  if (d_b) (void)hipFree(d_b);
  if (d_ranks) (void)hipFree(d_ranks);
  d_c = nullptr; d_r = nullptr; d_b = nullptr; d_ranks = nullptr;
}

// Build or fetch the cached per-call unit plan (plan.cpp builders;
// deterministic across ranks for a given key).
Engine::Plan& Engine::get_plan(int prim, long elems, Dtype dt, RedOp op,
                               uint64_t active_mask, int root) {
  PlanKey key{prim, elems, (int)dt, (int)op, active_mask, root};
  auto it = plans_.find(key);
  if (it != plans_.end()) return it->second;

  const int esize = dtype_size(dt);
  PlanData pd;
  switch (prim) {
    case 0:  // allreduce
      pd = build_plan(shape_, rank_, elems, esize, chunk_bytes_, active_mask,
                      slice_weights_);
      break;
    case 1:  // reduce
      pd = build_reduce_plan(shape_, rank_, root, elems, esize, chunk_bytes_,
                             active_mask, slice_weights_);
      break;
    case 2:  // broadcast
      pd = build_broadcast_plan(world_, rank_, root, elems, esize, chunk_bytes_);
      break;
    case 3:  // allgather (elems = per-rank in elems)
      pd = build_allgather_plan(world_, rank_, elems, esize, chunk_bytes_);
      break;
    case 4:  // alltoall (elems = per-rank slice elems)
      pd = build_alltoall_plan(world_, rank_, elems, esize, chunk_bytes_);
      break;
    case 5:  // reducescatter (elems = out elems)
      pd = build_reducescatter_plan(world_, rank_, elems, esize, chunk_bytes_,
                                    active_mask);
      break;
    default:
      throw std::runtime_error("unknown primitive");
  }

  Plan plan;
  plan.total_elems = elems;
  plan.dt = dt;
  plan.cunits = std::move(pd.cunits);
  plan.runits = std::move(pd.runits);
  plan.bunits = std::move(pd.bunits);

  // End-barrier WAIT set: the call's active set for the masked
  // primitives, so a wedged excluded straggler cannot stall an active
  // call until the timeout (round-1 verdict item 8); everyone for the
  // unmasked primitives. The barrier kernel SIGNALS done to every rank
  // regardless, so an alive excluded relay that calls (to receive the
  // result) still synchronizes; if it lags far enough that a publisher
  // reused its buffer, the bcast pull's staleness re-check raises
  // instead of reading torn data.
  std::vector<int> ranks;
  if (prim == 0 || prim == 1 || prim == 5) {
    for (int r = 0; r < world_; ++r)
      if ((active_mask >> r) & 1) ranks.push_back(r);
  } else {
    for (int r = 0; r < world_; ++r) ranks.push_back(r);
  }
  plan.nranks = (int)ranks.size();

  // upload
  auto upload = [](const void* src, size_t bytes, void** dst) {
    if (!bytes) return;
    HIP_CHECK(hipMalloc(dst, bytes));
    HIP_CHECK(hipMemcpy(*dst, src, bytes, hipMemcpyHostToDevice));
  };
  upload(plan.cunits.data(), plan.cunits.size() * sizeof(CopyUnit),
         reinterpret_cast<void**>(&plan.d_c));
  upload(plan.runits.data(), plan.runits.size() * sizeof(ReduceUnit),
         reinterpret_cast<void**>(&plan.d_r));
  upload(plan.bunits.data(), plan.bunits.size() * sizeof(BcastUnit),
         reinterpret_cast<void**>(&plan.d_b));
  upload(ranks.data(), ranks.size() * sizeof(int),
         reinterpret_cast<void**>(&plan.d_ranks));

  auto res = plans_.emplace(key, std::move(plan));
  return res.first->second;
}

uint64_t Engine::resolve_mask(const std::vector<int>& active_ranks) const {
  uint64_t mask = 0;
  if (active_ranks.empty()) {
    mask = (world_ >= 64) ? ~0ull : ((1ull << world_) - 1);
  } else {
    for (int r : active_ranks) {
      if (r < 0 || r >= world_)
        throw std::runtime_error("active rank " + std::to_string(r) +
                                 " out of range [0, " +
                                 std::to_string(world_) + ")");
      mask |= (1ull << r);
    }
  }
  if (mask == 0) throw std::runtime_error("empty active set");
  return mask;
}

void Engine::enqueue(const Plan& plan, const void* in, void* out,
                     CallArgs& args, void* caller_stream) {
  hipStream_t caller = reinterpret_cast<hipStream_t>(caller_stream);
  HIP_CHECK(hipSetDevice(device_));
  // Slot-alternated calls: call k reuses slot k % n_slots, so it only
  // serializes behind call k - n_slots (whose end barrier proved every
  // peer finished with that slot). With n_slots == 2, call k+1's
  // copy/reduce overlaps call k's broadcast+barrier on the other stream.
  const int slot = (int)(args.seq % (uint64_t)n_slots_);
  const DevTables& tabs = tabs_[slot];
  unsigned long long* ctr = counters_ + (size_t)slot * 3 * kMaxUnits;
  if (args.seq > (uint64_t)n_slots_)
    HIP_CHECK(hipStreamWaitEvent(s_red_, ev_barrier_[slot], 0));
  HIP_CHECK(hipEventRecord(ev_in_, caller));
  HIP_CHECK(hipStreamWaitEvent(s_red_, ev_in_, 0));

  // Small-message fast path: one fused kernel on one stream replaces the
  // 4-launch pipeline (+ counter memsets + cross-stream events).
  if (args.total_elems * dtype_size(args.dtype) <= small_fused_bytes_) {
    launch_small_fused(args.dtype, in, out, plan.d_c, (int)plan.cunits.size(),
                       plan.d_r, (int)plan.runits.size(), plan.d_b,
                       (int)plan.bunits.size(), tabs, args, rank_, world_,
                       plan.d_ranks, plan.nranks, s_red_);
    HIP_CHECK(hipGetLastError());
    HIP_CHECK(hipEventRecord(ev_bc_, s_red_));
    HIP_CHECK(hipEventRecord(ev_barrier_[slot], s_red_));
    HIP_CHECK(hipStreamWaitEvent(caller, ev_bc_, 0));
    return;
  }

  const size_t cu64 = sizeof(unsigned long long);
  if (!plan.cunits.empty())
    HIP_CHECK(hipMemsetAsync(ctr, 0, plan.cunits.size() * cu64, s_red_));
  if (!plan.runits.empty())
    HIP_CHECK(hipMemsetAsync(ctr + kMaxUnits, 0,
                             plan.runits.size() * cu64, s_red_));
  if (!plan.bunits.empty())
    HIP_CHECK(hipMemsetAsync(ctr + 2 * kMaxUnits, 0,
                             plan.bunits.size() * cu64, s_red_));
  HIP_CHECK(hipEventRecord(ev_sync0_, s_red_));
  HIP_CHECK(hipStreamWaitEvent(s_bcast_, ev_sync0_, 0));

  DevTables launch_tabs = tabs;
  launch_tabs.counters = ctr;
  launch_collective(args.dtype, in, out, plan.d_c, (int)plan.cunits.size(),
                    plan.d_r, (int)plan.runits.size(), plan.d_b,
                    (int)plan.bunits.size(), launch_tabs, args, rank_,
                    ctr + kMaxUnits, ctr + 2 * kMaxUnits,
                    wgs_per_group_, n_groups_, s_red_, s_bcast_);
  HIP_CHECK(hipGetLastError());

  // caller sees "result ready" (bcast wrote the user tensor); the end
  // barrier only gates the engine's own next same-slot call on s_red
  HIP_CHECK(hipEventRecord(ev_bc_, s_bcast_));
  HIP_CHECK(hipEventRecord(ev_red_, s_red_));
  HIP_CHECK(hipStreamWaitEvent(s_bcast_, ev_red_, 0));
  launch_barrier(launch_tabs, args, rank_, world_, plan.d_ranks, plan.nranks,
                 s_bcast_);
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipEventRecord(ev_barrier_[slot], s_bcast_));
  HIP_CHECK(hipStreamWaitEvent(caller, ev_bc_, 0));
}

CallArgs Engine::make_args(Dtype dt, RedOp op, float scale, long elems) {
  CallArgs args{};
  args.seq = ++seq_;
  args.dtype = dt;
  args.op = op;
  args.scale = scale;
  args.total_elems = elems;
  args.timeout_ticks = (uint64_t)(timeout_ms_ * 100000.0);
  return args;
}

void Engine::check_ready(long bytes_needed) const {
  if (!connected_ && world_ > 1)
    throw std::runtime_error("engine not connected");
  if ((size_t)bytes_needed > cap_bytes_)
    throw std::runtime_error("tensor larger than engine capacity; split the call");
}

void Engine::allreduce(void* data, long total_elems, int dtype, int op,
                       const std::vector<int>& active_ranks, bool average,
                       void* caller_stream) {
  if (world_ == 1) return;
  Dtype dt = (Dtype)dtype;
  RedOp rop = (RedOp)op;
  check_ready(total_elems * dtype_size(dt));
  uint64_t mask = resolve_mask(active_ranks);
  int n_active = __builtin_popcountll(mask);
  Plan& plan = get_plan(0, total_elems, dt, rop, mask, -1);
  CallArgs args = make_args(dt, rop, (rop == RedOp::Avg) ? 1.0f / n_active : 1.0f,
                            total_elems);
  enqueue(plan, data, data, args, caller_stream);
}

void Engine::reduce(void* data, long total_elems, int dtype, int op, int root,
                    const std::vector<int>& active_ranks, void* caller_stream) {
  if (world_ == 1) return;
  Dtype dt = (Dtype)dtype;
  RedOp rop = (RedOp)op;
  check_ready(total_elems * dtype_size(dt));
  uint64_t mask = resolve_mask(active_ranks);
  int n_active = __builtin_popcountll(mask);
  Plan& plan = get_plan(1, total_elems, dt, rop, mask, root);
  CallArgs args = make_args(dt, rop, (rop == RedOp::Avg) ? 1.0f / n_active : 1.0f,
                            total_elems);
  enqueue(plan, data, data, args, caller_stream);
}

void Engine::broadcast(void* data, long total_elems, int dtype, int root,
                       void* caller_stream) {
  if (world_ == 1) return;
  Dtype dt = (Dtype)dtype;
  check_ready(total_elems * dtype_size(dt));
  Plan& plan = get_plan(2, total_elems, dt, RedOp::Sum,
                        (1ull << world_) - 1, root);
  CallArgs args = make_args(dt, RedOp::Sum, 1.0f, total_elems);
  enqueue(plan, data, data, args, caller_stream);
}

void Engine::all_gather(const void* in, void* out, long in_elems, int dtype,
                        void* caller_stream) {
  Dtype dt = (Dtype)dtype;
  check_ready((long)world_ * in_elems * dtype_size(dt));
  if (world_ == 1) {
    HIP_CHECK(hipMemcpyAsync(out, in, in_elems * dtype_size(dt),
                             hipMemcpyDeviceToDevice,
                             reinterpret_cast<hipStream_t>(caller_stream)));
    return;
  }
  Plan& plan = get_plan(3, in_elems, dt, RedOp::Sum, (1ull << world_) - 1, -1);
  CallArgs args = make_args(dt, RedOp::Sum, 1.0f, in_elems);
  enqueue(plan, in, out, args, caller_stream);
}

void Engine::all_to_all(const void* in, void* out, long per_rank_elems,
                        int dtype, void* caller_stream) {
  Dtype dt = (Dtype)dtype;
  check_ready((long)world_ * per_rank_elems * dtype_size(dt));
  if (world_ == 1) {
    HIP_CHECK(hipMemcpyAsync(out, in, per_rank_elems * dtype_size(dt),
                             hipMemcpyDeviceToDevice,
                             reinterpret_cast<hipStream_t>(caller_stream)));
    return;
  }
  Plan& plan = get_plan(4, per_rank_elems, dt, RedOp::Sum,
                        (1ull << world_) - 1, -1);
  CallArgs args = make_args(dt, RedOp::Sum, 1.0f, per_rank_elems);
  enqueue(plan, in, out, args, caller_stream);
}

void Engine::reduce_scatter(const void* in, void* out, long out_elems,
                            int dtype, int op,
                            const std::vector<int>& active_ranks, bool average,
                            void* caller_stream) {
  Dtype dt = (Dtype)dtype;
  RedOp rop = (RedOp)op;
  check_ready((long)world_ * out_elems * dtype_size(dt));
  if (world_ == 1) {
    HIP_CHECK(hipMemcpyAsync(out, in, out_elems * dtype_size(dt),
                             hipMemcpyDeviceToDevice,
                             reinterpret_cast<hipStream_t>(caller_stream)));
    return;
  }
  uint64_t mask = resolve_mask(active_ranks);
  int n_active = __builtin_popcountll(mask);
  Plan& plan = get_plan(5, out_elems, dt, rop, mask, -1);
  CallArgs args = make_args(dt, rop,
                            (average || rop == RedOp::Avg) ? 1.0f / n_active : 1.0f,
                            out_elems);
  enqueue(plan, in, out, args, caller_stream);
}

std::string Engine::dump_inbox() {
  std::string out(sizeof(FlagInbox), '\0');
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipMemcpy(out.data(), tabs_[0].inbox[rank_], sizeof(FlagInbox),
                      hipMemcpyDeviceToHost));
  return out;
}

std::pair<uint64_t, uint64_t> Engine::query_error() {
  if (world_ == 1 || !connected_) return {0, 0};
  FlagInbox* inbox = tabs_[0].inbox[rank_];
  // pipelined mode: check both slots' error words
  if (n_slots_ > 1) {
    HIP_CHECK(hipMemcpyAsync(h_err_, &tabs_[1].inbox[rank_]->error,
                             2 * sizeof(uint64_t), hipMemcpyDeviceToHost,
                             s_err_));
    HIP_CHECK(hipStreamSynchronize(s_err_));
    if (h_err_[0] != 0) return {h_err_[0], h_err_[1]};
  }
  HIP_CHECK(hipMemcpyAsync(h_err_, &inbox->error, 2 * sizeof(uint64_t),
                           hipMemcpyDeviceToHost, s_err_));
  HIP_CHECK(hipStreamSynchronize(s_err_));
  return {h_err_[0], h_err_[1]};
}

void Engine::synchronize() {
  HIP_CHECK(hipStreamSynchronize(s_red_));
  HIP_CHECK(hipStreamSynchronize(s_bcast_));
  auto err = query_error();
  if (err.first != 0) {
    throw std::runtime_error(
        "adapcc engine kernel error code=" + std::to_string(err.first) +
        " detail=" + std::to_string(err.second) +
        " (peer timeout or wedged rank)");
  }
}

}  // namespace adapcc
