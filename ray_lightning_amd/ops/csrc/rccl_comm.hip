// Native RCCL communicator for the gradient data plane.
//
// Owns what the reference delegated to torch.distributed/NCCL
// (SURVEY.md N1/N3/N4): communicator bootstrap from an ncclUniqueId the
// Python layer hands out through the framework's control plane (no TCP
// store), and collectives launched on a DEDICATED HIP side stream with
// event-based ordering against the caller's compute stream — gradient
// all-reduce overlaps backward without host synchronization.
//
// Stream choreography per collective:
//   caller stream --record ev_in--> comm stream waits ev_in
//   ncclX(..., comm_stream); record ev_out[i] on comm stream
//   later: consumer stream waits ev_out[i]  (no host block)
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <chrono>
#include <cstdlib>
#include <memory>
#include <string>
#include <thread>
#include <vector>

#define CHECK_HIP(cmd)                                                     \
  do {                                                                     \
    hipError_t err__ = (cmd);                                              \
    TORCH_CHECK(err__ == hipSuccess, "HIP error: ",                        \
                hipGetErrorString(err__));                                 \
  } while (0)

#define CHECK_NCCL(cmd)                                                    \
  do {                                                                     \
    ncclResult_t res__ = (cmd);                                            \
    TORCH_CHECK(res__ == ncclSuccess, "RCCL error: ",                      \
                ncclGetErrorString(res__));                                \
  } while (0)

namespace {

constexpr int kEventPool = 512;

struct RcclComm {
  ncclComm_t comm = nullptr;
  hipStream_t stream = nullptr;     // dedicated comm side stream
  std::vector<hipEvent_t> events;   // ring of completion events
  hipEvent_t ev_in = nullptr;       // reusable input-dependency event
  long next_event = 0;
  int rank = 0;
  int world = 0;

  ~RcclComm() {
    if (comm) ncclCommDestroy(comm);
    for (auto e : events)
      if (e) (void)hipEventDestroy(e);
    if (ev_in) (void)hipEventDestroy(ev_in);
    if (stream) (void)hipStreamDestroy(stream);
  }
};

std::vector<std::shared_ptr<RcclComm>> g_comms;

ncclDataType_t to_nccl_dtype(const torch::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return ncclFloat32;
    case at::kBFloat16: return ncclBfloat16;
    case at::kHalf: return ncclFloat16;
    case at::kDouble: return ncclFloat64;
    case at::kInt: return ncclInt32;
    case at::kLong: return ncclInt64;
    case at::kByte: return ncclUint8;
    default:
      TORCH_CHECK(false, "unsupported dtype for RCCL collective");
  }
}

ncclRedOp_t to_nccl_op(const std::string& op) {
  if (op == "sum") return ncclSum;
  if (op == "max") return ncclMax;
  if (op == "min") return ncclMin;
  if (op == "product") return ncclProd;
  TORCH_CHECK(false, "unsupported reduce op: ", op);
}

RcclComm& get(long handle) {
  TORCH_CHECK(handle >= 0 && handle < (long)g_comms.size() &&
              g_comms[handle], "invalid comm handle");
  return *g_comms[handle];
}

// Surface asynchronous RCCL failures (peer death, xGMI link error)
// before enqueuing more work on a poisoned communicator — otherwise the
// next collective hangs the rank instead of raising.
void check_async(RcclComm& c) {
  ncclResult_t st = ncclSuccess;
  ncclResult_t q = ncclCommGetAsyncError(c.comm, &st);
  TORCH_CHECK(q == ncclSuccess, "ncclCommGetAsyncError failed: ",
              ncclGetErrorString(q));
  TORCH_CHECK(st == ncclSuccess || st == ncclInProgress,
              "RCCL communicator (rank ", c.rank, "/", c.world,
              ") is in error state: ", ncclGetErrorString(st));
}

// The communicator is non-blocking (see comm_init): a collective may
// return ncclInProgress while a background thread finishes posting it
// to the stream. The completion event must NOT be recorded until the
// op is actually enqueued, or consumers could wait on an event that
// precedes the collective. Poll until posted (normally instantaneous
// once connections exist; bounded for safety).
void wait_posted(RcclComm& c, ncclResult_t res) {
  if (res == ncclSuccess) return;
  TORCH_CHECK(res == ncclInProgress, "RCCL collective failed (rank ",
              c.rank, "/", c.world, "): ", ncclGetErrorString(res));
  auto deadline =
      std::chrono::steady_clock::now() + std::chrono::seconds(300);
  ncclResult_t st = ncclInProgress;
  while (true) {
    ncclResult_t q = ncclCommGetAsyncError(c.comm, &st);
    TORCH_CHECK(q == ncclSuccess, "ncclCommGetAsyncError: ",
                ncclGetErrorString(q));
    if (st == ncclSuccess) return;
    TORCH_CHECK(st == ncclInProgress, "RCCL collective failed (rank ",
                c.rank, "/", c.world, "): ", ncclGetErrorString(st));
    TORCH_CHECK(std::chrono::steady_clock::now() < deadline,
                "RCCL collective enqueue stalled >300 s (rank ", c.rank,
                "/", c.world, ")");
    std::this_thread::yield();
  }
}

// Make comm stream wait on the caller's stream, run fn, record a
// completion event; returns the event index.
//
// Event-pool wraparound note: kEventPool completion events are reused
// round-robin. Re-recording a slot a consumer has not waited on yet is
// safe — hipStreamWaitEvent issued later simply waits on the NEWER
// recording (over-synchronization, never under-) — and a DDP step
// issues ~#buckets collectives with waits every step, far below 512
// outstanding.
template <typename Fn>
long run_on_comm_stream(RcclComm& c, uintptr_t caller_stream, Fn&& fn) {
  check_async(c);
  hipStream_t cs = reinterpret_cast<hipStream_t>(caller_stream);
  CHECK_HIP(hipEventRecord(c.ev_in, cs));
  CHECK_HIP(hipStreamWaitEvent(c.stream, c.ev_in, 0));
  wait_posted(c, fn(c.stream));
  long idx = c.next_event++ % kEventPool;
  CHECK_HIP(hipEventRecord(c.events[idx], c.stream));
  return idx;
}

}  // namespace

py::bytes get_unique_id() {
  ncclUniqueId id;
  CHECK_NCCL(ncclGetUniqueId(&id));
  return py::bytes(reinterpret_cast<const char*>(&id), sizeof(id));
}

long comm_init(py::bytes uid_bytes, long rank, long world) {
  std::string raw = uid_bytes;
  TORCH_CHECK(raw.size() == sizeof(ncclUniqueId),
              "uniqueId must be ", sizeof(ncclUniqueId), " bytes");
  ncclUniqueId id;
  memcpy(&id, raw.data(), sizeof(id));
  auto c = std::make_shared<RcclComm>();
  c->rank = (int)rank;
  c->world = (int)world;
  CHECK_HIP(hipStreamCreateWithFlags(&c->stream, hipStreamNonBlocking));
  c->events.resize(kEventPool);
  for (auto& e : c->events)
    CHECK_HIP(hipEventCreateWithFlags(&e, hipEventDisableTiming));
  CHECK_HIP(hipEventCreateWithFlags(&c->ev_in, hipEventDisableTiming));

  // Non-blocking init with a deadline: a blocking ncclCommInitRank hangs
  // the whole job forever if any peer died between uniqueId handout and
  // init (the classic multi-rank bringup failure). Poll the async state
  // and abort with a diagnosable error instead.
  double timeout_s = 180.0;
  if (const char* env = std::getenv("RLA_RCCL_INIT_TIMEOUT_S")) {
    timeout_s = std::atof(env);
    if (timeout_s <= 0) timeout_s = 180.0;
  }
  ncclConfig_t config = NCCL_CONFIG_INITIALIZER;
  config.blocking = 0;
  ncclResult_t res =
      ncclCommInitRankConfig(&c->comm, (int)world, id, (int)rank, &config);
  TORCH_CHECK(res == ncclSuccess || res == ncclInProgress,
              "ncclCommInitRankConfig: ", ncclGetErrorString(res));
  auto deadline = std::chrono::steady_clock::now() +
                  std::chrono::duration<double>(timeout_s);
  ncclResult_t st = ncclInProgress;
  while (true) {
    CHECK_NCCL(ncclCommGetAsyncError(c->comm, &st));
    if (st == ncclSuccess) break;
    TORCH_CHECK(st == ncclInProgress, "RCCL init failed (rank ", rank,
                "/", world, "): ", ncclGetErrorString(st));
    if (std::chrono::steady_clock::now() >= deadline) {
      (void)ncclCommAbort(c->comm);
      c->comm = nullptr;  // abort already destroyed it
      TORCH_CHECK(false, "RCCL init timed out after ", timeout_s,
                  " s (rank ", rank, "/", world,
                  "): a peer rank likely died before ncclCommInitRank; "
                  "set RLA_RCCL_INIT_TIMEOUT_S to adjust");
    }
    std::this_thread::sleep_for(std::chrono::milliseconds(5));
  }
  g_comms.push_back(c);
  return (long)g_comms.size() - 1;
}

long comm_rank(long h) { return get(h).rank; }
long comm_world_size(long h) { return get(h).world; }

long all_reduce(long h, torch::Tensor t, std::string op,
                uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    return ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                         to_nccl_dtype(t), to_nccl_op(op), c.comm, s);
  });
}

long broadcast(long h, torch::Tensor t, long root,
               uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    return ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                         to_nccl_dtype(t), (int)root, c.comm, s);
  });
}

long reduce(long h, torch::Tensor t, long dst, std::string op,
            uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    return ncclReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                      to_nccl_dtype(t), to_nccl_op(op), (int)dst,
                      c.comm, s);
  });
}

long reduce_scatter(long h, torch::Tensor out, torch::Tensor in,
                    uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(out.is_cuda() && in.is_cuda());
  TORCH_CHECK(in.numel() == out.numel() * c.world,
              "reduce_scatter: input must be world_size * output");
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    return ncclReduceScatter(in.data_ptr(), out.data_ptr(), out.numel(),
                             to_nccl_dtype(in), ncclSum, c.comm, s);
  });
}

long all_gather(long h, torch::Tensor out, torch::Tensor in,
                uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(out.is_cuda() && in.is_cuda());
  TORCH_CHECK(out.numel() == in.numel() * c.world,
              "all_gather: output must be world_size * input");
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    return ncclAllGather(in.data_ptr(), out.data_ptr(), in.numel(),
                         to_nccl_dtype(in), c.comm, s);
  });
}

void stream_wait_event(long h, uintptr_t stream, long event_idx) {
  auto& c = get(h);
  TORCH_CHECK(event_idx >= 0 && event_idx < kEventPool);
  CHECK_HIP(hipStreamWaitEvent(reinterpret_cast<hipStream_t>(stream),
                               c.events[event_idx], 0));
}

void comm_stream_sync(long h) {
  CHECK_HIP(hipStreamSynchronize(get(h).stream));
}

void comm_destroy(long h) {
  if (h >= 0 && h < (long)g_comms.size()) g_comms[h].reset();
}

// Abort tears down the communicator without waiting for pending
// operations — the fail-fast path when a peer rank is known dead.
void comm_abort(long h) {
  if (h < 0 || h >= (long)g_comms.size() || !g_comms[h]) return;
  auto& c = *g_comms[h];
  if (c.comm) {
    (void)ncclCommAbort(c.comm);
    c.comm = nullptr;
  }
  g_comms[h].reset();
}

// grouped all-reduce of several tensors in one RCCL group call
long all_reduce_coalesced(long h, std::vector<torch::Tensor> ts,
                          std::string op, uintptr_t caller_stream) {
  auto& c = get(h);
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclGroupStart());
    for (auto& t : ts) {
      ncclResult_t r =
          ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                        to_nccl_dtype(t), to_nccl_op(op), c.comm, s);
      TORCH_CHECK(r == ncclSuccess || r == ncclInProgress,
                  "RCCL error: ", ncclGetErrorString(r));
    }
    return ncclGroupEnd();
  });
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("get_unique_id", &get_unique_id);
  m.def("comm_init", &comm_init);
  m.def("comm_rank", &comm_rank);
  m.def("comm_world_size", &comm_world_size);
  m.def("all_reduce", &all_reduce);
  m.def("all_reduce_coalesced", &all_reduce_coalesced);
  m.def("broadcast", &broadcast);
  m.def("reduce", &reduce);
  m.def("reduce_scatter", &reduce_scatter);
  m.def("all_gather", &all_gather);
  m.def("stream_wait_event", &stream_wait_event);
  m.def("comm_stream_sync", &comm_stream_sync);
  m.def("comm_destroy", &comm_destroy);
  m.def("comm_abort", &comm_abort);
}
