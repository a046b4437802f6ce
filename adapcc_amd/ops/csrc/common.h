// Common types for the adapcc_amd native engine.
//
// MI355X-native redesign of the reference's data plane (reference:
// csrc/include/init.h, csrc/include/trans.h). Differences by design:
//  - pull model: reducers read peer GPU memory directly over xGMI from
//    kernels (no staging copies, no cudaMemcpyPeerAsync chains, no MPI)
//  - flag "inboxes": producers push 8-byte system-scope release stores into
//    the consumer's local memory; consumers poll locally (no cross-link
//    polling traffic, no SysV-shm racy bools as in reference trans.cu:74)
//  - all spins are bounded by a hardware-clock timeout; a timed-out kernel
//    aborts and raises on the host instead of hanging the GPU
#pragma once

#include <cstdint>

namespace adapcc {

constexpr int kMaxRanks = 16;       // single node; 8 on MI355X (ref MAX_DEVICES=16)
constexpr int kMaxTrees = 16;       // parallel trees (ref MAX_TRANS=8)
constexpr int kMaxChunkSlots = 512; // per-tree chunk flag slots (ref MAX_CHUNK_NUM=512)
constexpr int kMaxSrcs = 16;        // max pull sources per reduce unit

// Which peer buffer a pull source refers to.
enum class BufKind : uint8_t { Send = 0, Acc = 1, Result = 2 };

enum class RedOp : uint8_t { Sum = 0, Avg = 1, Max = 2, Min = 3, PreScaleSum = 4 };

enum class Dtype : uint8_t { F32 = 0, F16 = 1, BF16 = 2 };

inline int dtype_size(Dtype d) { return d == Dtype::F32 ? 4 : 2; }

// ---------------------------------------------------------------------------
// Flag inbox layout (all u64 sequence values, GTE-waited).
// Each rank owns one inbox region in its comm buffer; PEERS write into it
// with system-scope release stores over xGMI; the owner polls locally.
//   ready[src][tree][slot] : src's (t,slot) contribution is pullable
//   bcast[tree][slot]      : this rank's tree-t parent published (t,slot)
//   done[src]              : src finished the current call (end barrier)
// ---------------------------------------------------------------------------
struct FlagInbox {
  uint64_t ready[kMaxRanks][kMaxTrees][kMaxChunkSlots];
  uint64_t bcast[kMaxTrees][kMaxChunkSlots];
  uint64_t done[kMaxRanks];
  // kernel error/status word: 0 ok; else error code (host-polled)
  uint64_t error;
  uint64_t error_detail;
  // debug trace slots (written by kernels; host-dumped)
  uint64_t trace[32];
};

// Error codes written by kernels on bounded-spin timeout.
constexpr uint64_t kErrTimeoutReady = 1;
constexpr uint64_t kErrTimeoutBcast = 2;
constexpr uint64_t kErrTimeoutDone = 3;
constexpr uint64_t kErrTimeoutCount = 4;
constexpr uint64_t kErrStaleRead = 5;

// One reduce work unit: pull nsrc buffers for (tree, chunk) and accumulate
// into dst. Precomputed on the host per call from strategy + active set.
struct ReduceUnit {
  int tree;
  int chunk;          // chunk index within this tree's slice
  long offset_elems;  // element offset of this chunk within the tensor
  long count_elems;   // elements in this chunk
  int nsrc;
  int src_rank[kMaxSrcs];
  uint8_t src_kind[kMaxSrcs];   // BufKind
  uint8_t include_self;         // also add own sendbuf (this rank active)
  uint8_t notify_parent;        // push ready flag to parent after reduce
  int parent_rank;              // valid if notify_parent
  uint8_t is_root;              // publish bcast flags to children
  int nchildren;                // fanout ranks to notify on publish
  int child_rank[kMaxRanks];
};

// One copy-in unit: stage a user-tensor range into sendbuf at (t, chunk)
// and notify its consumers that it is pullable. flag_space selects which
// inbox array the notification lands in: 0 -> ready[me][t][c] (a reducer
// will pull), 1 -> bcast[t][c] (a receiver will pull, e.g. broadcast /
// allgather source).
struct CopyUnit {
  int tree;
  int chunk;
  long offset_elems;
  long count_elems;
  uint8_t flag_space;  // 0 = ready, 1 = bcast
  int nnotify;
  int notify_rank[kMaxRanks];
};

// One receive/pull unit: wait for the (t,chunk) publication in the local
// bcast inbox, pull count elems from the source buffer at src_offset into
// the user OUT tensor at dst_offset (scaled); forward: also write resultbuf
// at src_offset and re-publish to children (multi-level trees).
struct BcastUnit {
  int tree;
  int chunk;
  long src_offset_elems;
  long dst_offset_elems;
  long count_elems;
  int parent_rank;      // who to pull from (-1: self -> local buffer)
  uint8_t parent_kind;  // BufKind of the source buffer
  uint8_t forward;
  int nchildren;
  int child_rank[kMaxRanks];
};

// Peer buffer tables passed by value to kernels. send/acc/result are
// tensor-shaped regions (same element offsets as the user tensor); inbox is
// each rank's flag inbox; counters is local per-call arrival scratch.
struct DevTables {
  const void* send[kMaxRanks];
  void* acc[kMaxRanks];
  void* result[kMaxRanks];
  FlagInbox* inbox[kMaxRanks];
  unsigned long long* counters;
};

// Device-visible per-call arguments (buffers of units are device memory).
struct CallArgs {
  uint64_t seq;            // monotonically increasing call sequence
  Dtype dtype;
  RedOp op;
  float scale;             // applied on the broadcast write (e.g. 1/n_active)
  long total_elems;
  uint64_t timeout_ticks;  // s_memrealtime ticks (100 MHz) before abort
};

}  // namespace adapcc
