// Direct-RCCL bucket communication core (SURVEY §2.2 "Comm backend
// binding", §5.8).
//
// torch's ProcessGroupNCCL is used for init/broadcast/misc, but the DDP
// hot path — one SUM all-reduce per gradient bucket, overlapped with
// backward — goes through this thin C++ layer calling the RCCL C API
// directly, so bucket↔channel behavior (NCCL_MIN_NCHANNELS vs the 7
// xGMI links per MI355X GPU) and the exposed-comm time are under the
// framework's control and instrumentable.
//
// Stream/event design: one dedicated comm hipStream per communicator.
//   all_reduce(t):  event(E) on the CURRENT torch stream  ->
//                   comm stream waits E  -> ncclAllReduce on comm stream
//   join():         event(F) on comm stream -> current stream waits F
// No host syncs anywhere on the step path (C4).  Per-bucket event pairs
// on the comm stream accumulate measured collective time (metrics §5.5
// "comm-exposed").

#include <rccl/rccl.h>

#include <memory>
#include <vector>

#include "../common.h"

namespace {

#define RCCL_CHECK(expr)                                                      \
  do {                                                                        \
    ncclResult_t _r = (expr);                                                 \
    TORCH_CHECK(_r == ncclSuccess, "RCCL error: ", ncclGetErrorString(_r));   \
  } while (0)

struct RcclComm {
  ncclComm_t comm = nullptr;
  hipStream_t stream = nullptr;
  std::vector<std::pair<hipEvent_t, hipEvent_t>> timers;  // per-bucket
  size_t timers_used = 0;
  // pooled ordering events (hot path: one per all_reduce + one per join;
  // created once, reused — per-call create/destroy churned the runtime)
  std::vector<hipEvent_t> order_pool;
  size_t order_next = 0;
  int rank = -1, world = 0;

  hipEvent_t next_order_event() {
    if (order_pool.empty()) {
      order_pool.resize(64, nullptr);
      for (auto& e : order_pool)
        HIP_CHECK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
    }
    hipEvent_t e = order_pool[order_next];
    order_next = (order_next + 1) % order_pool.size();
    return e;
  }

  ~RcclComm() {
    for (auto& p : timers) {
      (void)hipEventDestroy(p.first);
      (void)hipEventDestroy(p.second);
    }
    for (auto& e : order_pool) (void)hipEventDestroy(e);
    if (stream) (void)hipStreamDestroy(stream);
    if (comm) (void)ncclCommDestroy(comm);
  }
};

std::vector<std::shared_ptr<RcclComm>> g_comms;

ncclDataType_t nccl_dtype(const torch::Tensor& t) {
  switch (t.scalar_type()) {
    case torch::kBFloat16: return ncclBfloat16;
    case torch::kFloat32: return ncclFloat32;
    case torch::kFloat16: return ncclFloat16;
    case torch::kInt64: return ncclInt64;
    case torch::kInt32: return ncclInt32;
    default: TORCH_CHECK(false, "rccl: unsupported dtype ", t.scalar_type());
  }
}

std::pair<hipEvent_t, hipEvent_t>& next_timer(RcclComm& c) {
  if (c.timers_used == c.timers.size()) {
    hipEvent_t a, b;
    HIP_CHECK(hipEventCreate(&a));
    HIP_CHECK(hipEventCreate(&b));
    c.timers.emplace_back(a, b);
  }
  return c.timers[c.timers_used++];
}

}  // namespace

// rank 0 calls this, broadcasts the bytes over the bootstrap PG
torch::Tensor rccl_unique_id() {
  ncclUniqueId id;
  RCCL_CHECK(ncclGetUniqueId(&id));
  auto t = torch::empty({(long)sizeof(ncclUniqueId)},
                        torch::TensorOptions().dtype(torch::kUInt8));
  memcpy(t.data_ptr(), &id, sizeof(id));
  return t;
}

long rccl_init(long rank, long world, torch::Tensor id_bytes) {
  TORCH_CHECK(id_bytes.numel() == (long)sizeof(ncclUniqueId));
  ncclUniqueId id;
  memcpy(&id, id_bytes.contiguous().data_ptr(), sizeof(id));
  auto c = std::make_shared<RcclComm>();
  c->rank = (int)rank;
  c->world = (int)world;
  HIP_CHECK(hipStreamCreateWithFlags(&c->stream, hipStreamNonBlocking));
  RCCL_CHECK(ncclCommInitRank(&c->comm, (int)world, id, (int)rank));
  g_comms.push_back(c);
  return (long)g_comms.size() - 1;
}

// SUM all-reduce of one bucket on the comm stream, ordered after the
// current (compute) stream; returns immediately.
void rccl_all_reduce(long handle, torch::Tensor t) {
  auto& c = *g_comms.at(handle);
  CHECK_CUDA(t);
  CHECK_CONTIG(t);
  hipStream_t cur = scaffold_stream();
  hipEvent_t ready = c.next_order_event();
  HIP_CHECK(hipEventRecord(ready, cur));
  HIP_CHECK(hipStreamWaitEvent(c.stream, ready, 0));
  auto& tm = next_timer(c);
  HIP_CHECK(hipEventRecord(tm.first, c.stream));
  RCCL_CHECK(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                           nccl_dtype(t), ncclSum, c.comm, c.stream));
  HIP_CHECK(hipEventRecord(tm.second, c.stream));
}

// end-of-backward: compute stream waits for the comm stream (no host sync)
void rccl_join(long handle) {
  auto& c = *g_comms.at(handle);
  hipStream_t cur = scaffold_stream();
  hipEvent_t done = c.next_order_event();
  HIP_CHECK(hipEventRecord(done, c.stream));
  HIP_CHECK(hipStreamWaitEvent(cur, done, 0));
}

// total measured collective time (ms) since last call; host-syncs the
// recorded events, so call OUTSIDE the timed region (log cadence only)
double rccl_comm_time_ms(long handle) {
  auto& c = *g_comms.at(handle);
  double total = 0.0;
  for (size_t i = 0; i < c.timers_used; ++i) {
    HIP_CHECK(hipEventSynchronize(c.timers[i].second));
    float ms = 0.f;
    HIP_CHECK(hipEventElapsedTime(&ms, c.timers[i].first, c.timers[i].second));
    total += ms;
  }
  c.timers_used = 0;
  return total;
}

void rccl_destroy(long handle) {
  g_comms.at(handle).reset();
}
