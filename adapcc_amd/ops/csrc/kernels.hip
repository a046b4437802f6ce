// adapcc_amd device kernels — CDNA4 (gfx950) native.
//
// The data plane is a *pull* pipeline: reducers read peer GPU memory
// directly over xGMI from inside kernels (hipIpc-mapped), synchronized by
// 8-byte system-scope release stores pushed into the consumer's local
// "inbox" (consumers poll local HBM/L2, never a remote link).
//
// Replaces the reference's host-threaded chunk loop
// (reference: csrc/allreduce.cu:430-666 reduce thread, csrc/trans.cu:10-24
// reduceSumKernel, csrc/trans.cu:58-100 cudaSend/cudaRecv spin machinery):
// here the GPU itself runs the per-chunk pipeline; the host only enqueues
// three kernels per collective. Every spin is bounded by a hardware-clock
// deadline; on timeout the kernel aborts and posts an error code the host
// raises from.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include "common.h"

namespace adapcc {

#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// Sync helpers (Guideline-16 shapes at system scope: peers are other GPUs)
// ---------------------------------------------------------------------------

DEV_INLINE unsigned long long realtime() {
  return __builtin_amdgcn_s_memrealtime();  // ~100 MHz constant clock
}

DEV_INLINE void push_flag(uint64_t* dst, uint64_t v) {
  // 8-byte system-scope release store (possibly over xGMI into a peer's
  // inbox). Prior data stores must already be drained by the caller.
  __hip_atomic_store(dst, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
}

DEV_INLINE uint64_t peek_flag(const uint64_t* f) {
  return __hip_atomic_load(f, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

DEV_INLINE void post_error(FlagInbox* my_inbox, uint64_t code, uint64_t detail) {
  __hip_atomic_store(&my_inbox->error_detail, detail, __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_SYSTEM);
  __hip_atomic_store(&my_inbox->error, code, __ATOMIC_RELEASE,
                     __HIP_MEMORY_SCOPE_SYSTEM);
}

// Whole workgroup waits until *flag >= seq (local poll by lane 0, broadcast
// via barrier, one system-acquire fence). Returns false on timeout.
DEV_INLINE bool wait_flag_ge(const uint64_t* flag, uint64_t seq,
                             unsigned long long deadline, FlagInbox* my_inbox,
                             uint64_t code, uint64_t detail) {
  __shared__ int ok_sh;
  if (threadIdx.x == 0) {
    int ok = 1;
    while (peek_flag(flag) < seq) {
      __builtin_amdgcn_s_sleep(32);
      if (realtime() > deadline) {
        ok = 0;
        post_error(my_inbox, code, detail);
        break;
      }
    }
    ok_sh = ok;
  }
  __syncthreads();
  const bool ok = ok_sh != 0;
  if (ok && threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");  // system scope
  }
  __syncthreads();
  return ok;
}

// Arrive at a per-unit counter after this workgroup's stores; returns true
// for the LAST arriving workgroup (which may then push flags: all other
// workgroups' data is globally visible at that point).
DEV_INLINE bool unit_arrive(unsigned long long* counter, unsigned expect) {
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every wave drains
  __syncthreads();
  if (expect == 1) {  // single workgroup: no counter needed (fused path)
    if (threadIdx.x == 0) {
      __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");  // system release
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __syncthreads();
    return true;
  }
  __shared__ int last_sh;
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");  // system release
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    unsigned long long prev = __hip_atomic_fetch_add(
        counter, 1ull, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
    last_sh = (prev == (unsigned long long)(expect - 1)) ? 1 : 0;
  }
  __syncthreads();
  return last_sh != 0;
}

// ---------------------------------------------------------------------------
// dtype plumbing: vectorized loads as 16-byte packets, accumulate in fp32
// ---------------------------------------------------------------------------

template <typename T> struct VecTraits;

template <> struct VecTraits<float> {
  static constexpr int kPerVec = 4;
  using Vec = float4;
  DEV_INLINE static void unpack(const Vec& v, float* out) {
    out[0] = v.x; out[1] = v.y; out[2] = v.z; out[3] = v.w;
  }
  DEV_INLINE static Vec pack(const float* in) {
    return make_float4(in[0], in[1], in[2], in[3]);
  }
  DEV_INLINE static float load1(const float* p) { return *p; }
  DEV_INLINE static void store1(float* p, float v) { *p = v; }
};

template <> struct VecTraits<__hip_bfloat16> {
  static constexpr int kPerVec = 8;
  struct Vec { uint4 raw; };
  DEV_INLINE static void unpack(const Vec& v, float* out) {
    const unsigned* w = reinterpret_cast<const unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat162 h = *reinterpret_cast<const __hip_bfloat162*>(&w[i]);
      out[2 * i] = __bfloat162float(h.x);
      out[2 * i + 1] = __bfloat162float(h.y);
    }
  }
  DEV_INLINE static Vec pack(const float* in) {
    Vec v;
    unsigned* w = reinterpret_cast<unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __hip_bfloat162 h{__float2bfloat16(in[2 * i]), __float2bfloat16(in[2 * i + 1])};
      w[i] = *reinterpret_cast<const unsigned*>(&h);
    }
    return v;
  }
  DEV_INLINE static float load1(const __hip_bfloat16* p) { return __bfloat162float(*p); }
  DEV_INLINE static void store1(__hip_bfloat16* p, float v) { *p = __float2bfloat16(v); }
};

template <> struct VecTraits<__half> {
  static constexpr int kPerVec = 8;
  struct Vec { uint4 raw; };
  DEV_INLINE static void unpack(const Vec& v, float* out) {
    const unsigned* w = reinterpret_cast<const unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __half2 h = *reinterpret_cast<const __half2*>(&w[i]);
      out[2 * i] = __half2float(__low2half(h));
      out[2 * i + 1] = __half2float(__high2half(h));
    }
  }
  DEV_INLINE static Vec pack(const float* in) {
    Vec v;
    unsigned* w = reinterpret_cast<unsigned*>(&v.raw);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      __half2 h = __floats2half2_rn(in[2 * i], in[2 * i + 1]);
      w[i] = *reinterpret_cast<const unsigned*>(&h);
    }
    return v;
  }
  DEV_INLINE static float load1(const __half* p) { return __half2float(*p); }
  DEV_INLINE static void store1(__half* p, float v) { *p = __float2half(v); }
};

DEV_INLINE float red_combine(RedOp op, float a, float b) {
  switch (op) {
    case RedOp::Max: return a > b ? a : b;
    case RedOp::Min: return a < b ? a : b;
    default: return a + b;  // Sum / Avg / PreScaleSum
  }
}

// ---------------------------------------------------------------------------
// Copy-in kernel: stage user tensor into sendbuf, chunk by chunk, pushing
// per-chunk readiness to the tree parent so remote reducers start pulling
// while later chunks are still staging.
// ---------------------------------------------------------------------------

template <typename T>
DEV_INLINE void copyin_body(
    const T* __restrict__ user, const CopyUnit* __restrict__ units, int n_units,
    const DevTables& tabs, const CallArgs& args, int me, int wgs_per_group,
    int n_groups, int group, int wg_in_group) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  T* sendbuf = (T*)tabs.send[me];

  for (int ui = group; ui < n_units; ui += n_groups) {
    const CopyUnit u = units[ui];
    const long cnt = u.count_elems;
    // this wg's share, rounded up to the vector width so every non-final
    // boundary stays 16-byte aligned
    const long per =
        ((cnt + wgs_per_group - 1) / wgs_per_group + VT::kPerVec - 1) /
        VT::kPerVec * VT::kPerVec;
    const long beg = min((long)wg_in_group * per, cnt);
    const long end = min(beg + per, cnt);
    const T* src = user + u.offset_elems;
    T* dst = sendbuf + u.offset_elems;

    long i = beg + (long)threadIdx.x * VT::kPerVec;
    const long vend = beg + ((end - beg) / VT::kPerVec) * VT::kPerVec;
    for (; i + VT::kPerVec <= vend; i += (long)blockDim.x * VT::kPerVec) {
      *reinterpret_cast<Vec*>(dst + i) = *reinterpret_cast<const Vec*>(src + i);
    }
    for (long j = vend + threadIdx.x; j < end; j += blockDim.x) {
      dst[j] = src[j];
    }

    if (u.nnotify > 0) {
      const bool last = unit_arrive(&tabs.counters[ui], wgs_per_group);
      if (last && threadIdx.x == 0) {
        for (int c = 0; c < u.nnotify; ++c) {
          FlagInbox* ib = tabs.inbox[u.notify_rank[c]];
          uint64_t* f = (u.flag_space == 0) ? &ib->ready[me][u.tree][u.chunk]
                                            : &ib->bcast[u.tree][u.chunk];
          push_flag(f, args.seq);
        }
      }
    }
  }
}

template <typename T>
__global__ void __launch_bounds__(256) copyin_kernel(
    const T* __restrict__ user, const CopyUnit* __restrict__ units, int n_units,
    DevTables tabs, CallArgs args, int me, int wgs_per_group, int n_groups) {
  copyin_body<T>(user, units, n_units, tabs, args, me, wgs_per_group,
                 n_groups, blockIdx.x / wgs_per_group,
                 blockIdx.x % wgs_per_group);
}

// ---------------------------------------------------------------------------
// Reduce kernel: per unit (tree, chunk) pull every effective source buffer
// over xGMI, accumulate in fp32, write local accbuf; push readiness up the
// tree; a root publishes the chunk to its broadcast fanout.
// ---------------------------------------------------------------------------

template <typename T>
DEV_INLINE void reduce_body(
    const ReduceUnit* __restrict__ units, int n_units, const DevTables& tabs,
    const CallArgs& args, int me, int wgs_per_group, int n_groups,
    unsigned long long* counters, int group, int wg_in_group) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  FlagInbox* my_inbox = tabs.inbox[me];
  const unsigned long long deadline = realtime() + args.timeout_ticks;

  for (int ui = group; ui < n_units; ui += n_groups) {
    const ReduceUnit u = units[ui];

    // Wait for every remote source's readiness flag (local poll).
    bool ok = true;
    for (int s = 0; s < u.nsrc && ok; ++s) {
      const int r = u.src_rank[s];
      if (r == me) continue;  // own sendbuf ordered by stream
      ok = wait_flag_ge(&my_inbox->ready[r][u.tree][u.chunk], args.seq, deadline,
                        my_inbox, kErrTimeoutReady,
                        ((uint64_t)u.tree << 32) | (uint32_t)u.chunk);
    }
    if (!ok) return;

    // Gather source pointers.
    const T* srcs[kMaxSrcs + 1];
    int ns = 0;
    for (int s = 0; s < u.nsrc; ++s) {
      const int r = u.src_rank[s];
      const void* base = (BufKind)u.src_kind[s] == BufKind::Send
                             ? tabs.send[r]
                             : (BufKind)u.src_kind[s] == BufKind::Acc
                                   ? (const void*)tabs.acc[r]
                                   : (const void*)tabs.result[r];
      srcs[ns++] = (const T*)base + u.offset_elems;
    }
    if (u.include_self) srcs[ns++] = (const T*)tabs.send[me] + u.offset_elems;

    T* dst = (T*)tabs.acc[me] + u.offset_elems;
    const long cnt = u.count_elems;
    const long per =
        ((cnt + wgs_per_group - 1) / wgs_per_group + VT::kPerVec - 1) /
        VT::kPerVec * VT::kPerVec;
    const long beg = min((long)wg_in_group * per, cnt);
    const long end = min(beg + per, cnt);
    const long vend = beg + ((end - beg) / VT::kPerVec) * VT::kPerVec;

    for (long i = beg + (long)threadIdx.x * VT::kPerVec; i + VT::kPerVec <= vend;
         i += (long)blockDim.x * VT::kPerVec) {
      float acc[VT::kPerVec];
      Vec v = *reinterpret_cast<const Vec*>(srcs[0] + i);
      VT::unpack(v, acc);
      for (int s = 1; s < ns; ++s) {
        float tmp[VT::kPerVec];
        Vec w = *reinterpret_cast<const Vec*>(srcs[s] + i);
        VT::unpack(w, tmp);
#pragma unroll
        for (int k = 0; k < VT::kPerVec; ++k) acc[k] = red_combine(args.op, acc[k], tmp[k]);
      }
      *reinterpret_cast<Vec*>(dst + i) = VT::pack(acc);
    }
    for (long j = vend + threadIdx.x; j < end; j += blockDim.x) {
      float a = VT::load1(srcs[0] + j);
      for (int s = 1; s < ns; ++s) a = red_combine(args.op, a, VT::load1(srcs[s] + j));
      VT::store1(dst + j, a);
    }

    // Arrive; last workgroup publishes.
    const bool last = unit_arrive(&counters[ui], wgs_per_group);
    if (last && threadIdx.x == 0) {
      if (u.notify_parent) {
        push_flag(&tabs.inbox[u.parent_rank]->ready[me][u.tree][u.chunk], args.seq);
      }
      if (u.is_root) {
        for (int c = 0; c < u.nchildren; ++c) {
          push_flag(&tabs.inbox[u.child_rank[c]]->bcast[u.tree][u.chunk], args.seq);
        }
      }
    }
  }
}

template <typename T>
__global__ void __launch_bounds__(256) reduce_kernel(
    const ReduceUnit* __restrict__ units, int n_units, DevTables tabs,
    CallArgs args, int me, int wgs_per_group, int n_groups,
    unsigned long long* counters /* = tabs.counters + copy units */) {
  reduce_body<T>(units, n_units, tabs, args, me, wgs_per_group, n_groups,
                 counters, blockIdx.x / wgs_per_group,
                 blockIdx.x % wgs_per_group);
}

// ---------------------------------------------------------------------------
// Broadcast/pull kernel: wait for the tree parent's publication, pull the
// chunk over xGMI into the user tensor (scaled); intermediates forward it
// (write resultbuf, publish to children).
// ---------------------------------------------------------------------------

template <typename T>
DEV_INLINE void bcast_body(
    T* __restrict__ user, const BcastUnit* __restrict__ units, int n_units,
    const DevTables& tabs, const CallArgs& args, int me, int wgs_per_group,
    int n_groups, unsigned long long* counters, int group, int wg_in_group) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  FlagInbox* my_inbox = tabs.inbox[me];
  const unsigned long long deadline = realtime() + args.timeout_ticks;
  const float scale = args.scale;

  for (int ui = group; ui < n_units; ui += n_groups) {
    const BcastUnit u = units[ui];
    if (!wait_flag_ge(&my_inbox->bcast[u.tree][u.chunk], args.seq, deadline,
                      my_inbox, kErrTimeoutBcast,
                      ((uint64_t)u.tree << 32) | (uint32_t)u.chunk)) {
      return;
    }
    const int src_rank = u.parent_rank < 0 ? me : u.parent_rank;
    const void* base = (BufKind)u.parent_kind == BufKind::Send
                           ? tabs.send[src_rank]
                           : (BufKind)u.parent_kind == BufKind::Acc
                                 ? (const void*)tabs.acc[src_rank]
                                 : (const void*)tabs.result[src_rank];
    const T* src = (const T*)base + u.src_offset_elems;
    T* dst = user + u.dst_offset_elems;
    T* fwd = u.forward ? (T*)tabs.result[me] + u.src_offset_elems : nullptr;

    const long cnt = u.count_elems;
    const long per =
        ((cnt + wgs_per_group - 1) / wgs_per_group + VT::kPerVec - 1) /
        VT::kPerVec * VT::kPerVec;
    const long beg = min((long)wg_in_group * per, cnt);
    const long end = min(beg + per, cnt);
    const long vend = beg + ((end - beg) / VT::kPerVec) * VT::kPerVec;

    for (long i = beg + (long)threadIdx.x * VT::kPerVec; i + VT::kPerVec <= vend;
         i += (long)blockDim.x * VT::kPerVec) {
      Vec v = *reinterpret_cast<const Vec*>(src + i);
      if (fwd) *reinterpret_cast<Vec*>(fwd + i) = v;  // unscaled forward
      float tmp[VT::kPerVec];
      VT::unpack(v, tmp);
#pragma unroll
      for (int k = 0; k < VT::kPerVec; ++k) tmp[k] *= scale;
      *reinterpret_cast<Vec*>(dst + i) = VT::pack(tmp);
    }
    for (long j = vend + threadIdx.x; j < end; j += blockDim.x) {
      float a = VT::load1(src + j);
      if (fwd) VT::store1(fwd + j, a);
      VT::store1(dst + j, a * scale);
    }

    // Staleness re-check: the end barrier only waits on the ACTIVE set,
    // so a far-lagging excluded reader could pull from a buffer its
    // parent already reused. The publish flag for this (tree, chunk)
    // slot advances past args.seq exactly when that happens — detect it
    // and raise instead of silently keeping torn data.
    if (threadIdx.x == 0 && u.parent_rank >= 0 && u.parent_rank != me) {
      if (peek_flag(&my_inbox->bcast[u.tree][u.chunk]) > args.seq) {
        post_error(my_inbox, kErrStaleRead,
                   ((uint64_t)u.tree << 32) | (uint32_t)u.chunk);
        return;
      }
    }

    if (u.forward) {
      const bool last = unit_arrive(&counters[ui], wgs_per_group);
      if (last && threadIdx.x == 0) {
        for (int c = 0; c < u.nchildren; ++c) {
          push_flag(&tabs.inbox[u.child_rank[c]]->bcast[u.tree][u.chunk], args.seq);
        }
      }
    }
  }
}

template <typename T>
__global__ void __launch_bounds__(256) bcast_kernel(
    T* __restrict__ user, const BcastUnit* __restrict__ units, int n_units,
    DevTables tabs, CallArgs args, int me, int wgs_per_group, int n_groups,
    unsigned long long* counters) {
  bcast_body<T>(user, units, n_units, tabs, args, me, wgs_per_group,
                n_groups, counters, blockIdx.x / wgs_per_group,
                blockIdx.x % wgs_per_group);
}

// ---------------------------------------------------------------------------
// End-of-call barrier kernel: publish "done reading peers" then wait for
// every peer's done — buffers are reusable for the next call after this.
// Runs after both the reduce and broadcast kernels (stream-ordered).
// ---------------------------------------------------------------------------

DEV_INLINE void barrier_body(const DevTables& tabs, const CallArgs& args,
                             int me, int world, const int* __restrict__ ranks,
                             int nranks) {
  FlagInbox* my_inbox = tabs.inbox[me];
  if (threadIdx.x == 0) {
    // signal EVERY rank (an excluded relay may be waiting to rejoin);
    // wait only on the plan's wait set below
    for (int r = 0; r < world; ++r) {
      if (r != me) push_flag(&tabs.inbox[r]->done[me], args.seq);
    }
  }
  __syncthreads();
  const unsigned long long deadline = realtime() + args.timeout_ticks;
  if (threadIdx.x < (unsigned)nranks) {
    const int r = ranks[threadIdx.x];
    if (r != me) {
      while (peek_flag(&my_inbox->done[r]) < args.seq) {
        __builtin_amdgcn_s_sleep(32);
        if (realtime() > deadline) {
          post_error(my_inbox, kErrTimeoutDone, (uint64_t)r);
          break;
        }
      }
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

__global__ void barrier_kernel(DevTables tabs, CallArgs args, int me,
                               int world, const int* __restrict__ ranks,
                               int nranks) {
  if (blockIdx.x != 0) return;
  barrier_body(tabs, args, me, world, ranks, nranks);
}

// ---------------------------------------------------------------------------
// Small-message fused collective: ONE workgroup, ONE launch runs the whole
// copy-in -> reduce -> broadcast -> end-barrier pipeline. For sub-256 KB
// buckets the 4-launch path is launch-latency-bound (~1.5 us per boundary
// + event syncs); this folds it into a single kernel. unit_arrive
// short-circuits at wgs_per_group == 1, so no counters are touched and no
// memset precedes the launch.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(256) small_fused_kernel(
    const T* __restrict__ user, T* __restrict__ user_mut,
    const CopyUnit* __restrict__ cunits, int nc,
    const ReduceUnit* __restrict__ runits, int nr,
    const BcastUnit* __restrict__ bunits, int nb, DevTables tabs,
    CallArgs args, int me, int world, const int* __restrict__ ranks,
    int nranks) {
  if (nc > 0) {
    copyin_body<T>(user, cunits, nc, tabs, args, me, 1, 1, 0, 0);
    __syncthreads();
  }
  if (nr > 0) {
    reduce_body<T>(runits, nr, tabs, args, me, 1, 1, nullptr, 0, 0);
    __syncthreads();
  }
  if (nb > 0) {
    bcast_body<T>(user_mut, bunits, nb, tabs, args, me, 1, 1, nullptr, 0, 0);
    __syncthreads();
  }
  barrier_body(tabs, args, me, world, ranks, nranks);
}

// ---------------------------------------------------------------------------
// Standalone multi-source reduction (no flags/tables): dst = op over nsrc
// local buffers. Used for single-GPU numerics tests of the reduction path
// and by fused ops (e.g. MoE combine). Same vectorized inner loop as the
// pull-reduce.
// ---------------------------------------------------------------------------

template <typename T>
__global__ void __launch_bounds__(256) local_reduce_kernel(
    T* __restrict__ dst, const T* const* __restrict__ srcs, int nsrc,
    long count, RedOp op, float scale) {
  using VT = VecTraits<T>;
  using Vec = typename VT::Vec;
  const long stride = (long)gridDim.x * blockDim.x;
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long vcount = count / VT::kPerVec;
  for (long i = tid; i < vcount; i += stride) {
    float acc[VT::kPerVec];
    Vec v = reinterpret_cast<const Vec*>(srcs[0])[i];
    VT::unpack(v, acc);
    for (int s = 1; s < nsrc; ++s) {
      float tmp[VT::kPerVec];
      Vec w = reinterpret_cast<const Vec*>(srcs[s])[i];
      VT::unpack(w, tmp);
#pragma unroll
      for (int k = 0; k < VT::kPerVec; ++k) acc[k] = red_combine(op, acc[k], tmp[k]);
    }
#pragma unroll
    for (int k = 0; k < VT::kPerVec; ++k) acc[k] *= scale;
    reinterpret_cast<Vec*>(dst)[i] = VT::pack(acc);
  }
  for (long j = vcount * VT::kPerVec + tid; j < count; j += stride) {
    float a = VT::load1(srcs[0] + j);
    for (int s = 1; s < nsrc; ++s) a = red_combine(op, a, VT::load1(srcs[s] + j));
    VT::store1(dst + j, a * scale);
  }
}

void launch_local_reduce(Dtype dt, void* dst, const void* const* srcs_dev,
                         int nsrc, long count, RedOp op, float scale,
                         hipStream_t stream) {
  const dim3 block(256);
  const dim3 grid(512);
  switch (dt) {
    case Dtype::F32:
      hipLaunchKernelGGL((local_reduce_kernel<float>), grid, block, 0, stream,
                         (float*)dst, (const float* const*)srcs_dev, nsrc, count,
                         op, scale);
      break;
    case Dtype::BF16:
      hipLaunchKernelGGL((local_reduce_kernel<__hip_bfloat16>), grid, block, 0,
                         stream, (__hip_bfloat16*)dst,
                         (const __hip_bfloat16* const*)srcs_dev, nsrc, count, op,
                         scale);
      break;
    case Dtype::F16:
      hipLaunchKernelGGL((local_reduce_kernel<__half>), grid, block, 0, stream,
                         (__half*)dst, (const __half* const*)srcs_dev, nsrc,
                         count, op, scale);
      break;
  }
}

// ---------------------------------------------------------------------------
// Launchers
// ---------------------------------------------------------------------------

template <typename T>
void launch_all(const void* user, void* user_mut, const CopyUnit* cunits, int nc,
                const ReduceUnit* runits, int nr, const BcastUnit* bunits, int nb,
                const DevTables& tabs, const CallArgs& args, int me,
                unsigned long long* red_counters, unsigned long long* bc_counters,
                int wgs_per_group, int n_groups, hipStream_t s_red,
                hipStream_t s_bcast) {
  const dim3 block(256);
  const dim3 grid(wgs_per_group * n_groups);
  if (nc > 0) {
    hipLaunchKernelGGL((copyin_kernel<T>), grid, block, 0, s_red, (const T*)user,
                       cunits, nc, tabs, args, me, wgs_per_group, n_groups);
  }
  if (nr > 0) {
    hipLaunchKernelGGL((reduce_kernel<T>), grid, block, 0, s_red, runits, nr, tabs,
                       args, me, wgs_per_group, n_groups, red_counters);
  }
  if (nb > 0) {
    hipLaunchKernelGGL((bcast_kernel<T>), grid, block, 0, s_bcast, (T*)user_mut,
                       bunits, nb, tabs, args, me, wgs_per_group, n_groups,
                       bc_counters);
  }
}

void launch_collective(Dtype dt, const void* user, void* user_mut,
                       const CopyUnit* cunits, int nc, const ReduceUnit* runits,
                       int nr, const BcastUnit* bunits, int nb,
                       const DevTables& tabs, const CallArgs& args, int me,
                       unsigned long long* red_counters,
                       unsigned long long* bc_counters, int wgs_per_group,
                       int n_groups, hipStream_t s_red, hipStream_t s_bcast) {
  switch (dt) {
    case Dtype::F32:
      launch_all<float>(user, user_mut, cunits, nc, runits, nr, bunits, nb, tabs,
                        args, me, red_counters, bc_counters, wgs_per_group,
                        n_groups, s_red, s_bcast);
      break;
    case Dtype::BF16:
      launch_all<__hip_bfloat16>(user, user_mut, cunits, nc, runits, nr, bunits,
                                 nb, tabs, args, me, red_counters, bc_counters,
                                 wgs_per_group, n_groups, s_red, s_bcast);
      break;
    case Dtype::F16:
      launch_all<__half>(user, user_mut, cunits, nc, runits, nr, bunits, nb, tabs,
                         args, me, red_counters, bc_counters, wgs_per_group,
                         n_groups, s_red, s_bcast);
      break;
  }
}

void launch_barrier(const DevTables& tabs, const CallArgs& args, int me, int world,
                    const int* ranks_dev, int nranks, hipStream_t stream) {
  hipLaunchKernelGGL(barrier_kernel, dim3(1), dim3(256), 0, stream, tabs, args, me,
                     world, ranks_dev, nranks);
}

void launch_small_fused(Dtype dt, const void* user, void* user_mut,
                        const CopyUnit* cunits, int nc,
                        const ReduceUnit* runits, int nr,
                        const BcastUnit* bunits, int nb, const DevTables& tabs,
                        const CallArgs& args, int me, int world,
                        const int* ranks_dev, int nranks, hipStream_t stream) {
  const dim3 grid(1), block(256);
  switch (dt) {
    case Dtype::F32:
      hipLaunchKernelGGL((small_fused_kernel<float>), grid, block, 0, stream,
                         (const float*)user, (float*)user_mut, cunits, nc,
                         runits, nr, bunits, nb, tabs, args, me, world,
                         ranks_dev, nranks);
      break;
    case Dtype::BF16:
      hipLaunchKernelGGL((small_fused_kernel<__hip_bfloat16>), grid, block, 0,
                         stream, (const __hip_bfloat16*)user,
                         (__hip_bfloat16*)user_mut, cunits, nc, runits, nr,
                         bunits, nb, tabs, args, me, world, ranks_dev, nranks);
      break;
    case Dtype::F16:
      hipLaunchKernelGGL((small_fused_kernel<__half>), grid, block, 0, stream,
                         (const __half*)user, (__half*)user_mut, cunits, nc,
                         runits, nr, bunits, nb, tabs, args, me, world,
                         ranks_dev, nranks);
      break;
  }
}

}  // namespace adapcc
