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

#include <memory>
#include <string>
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

// Make comm stream wait on the caller's stream, run fn, record a
// completion event; returns the event index.
template <typename Fn>
long run_on_comm_stream(RcclComm& c, uintptr_t caller_stream, Fn&& fn) {
  hipStream_t cs = reinterpret_cast<hipStream_t>(caller_stream);
  CHECK_HIP(hipEventRecord(c.ev_in, cs));
  CHECK_HIP(hipStreamWaitEvent(c.stream, c.ev_in, 0));
  fn(c.stream);
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
  CHECK_NCCL(ncclCommInitRank(&c->comm, (int)world, id, (int)rank));
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
    CHECK_NCCL(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                             to_nccl_dtype(t), to_nccl_op(op), c.comm,
                             s));
  });
}

long broadcast(long h, torch::Tensor t, long root,
               uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                             to_nccl_dtype(t), (int)root, c.comm, s));
  });
}

long reduce(long h, torch::Tensor t, long dst, std::string op,
            uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(t.is_cuda() && t.is_contiguous());
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                          to_nccl_dtype(t), to_nccl_op(op), (int)dst,
                          c.comm, s));
  });
}

long reduce_scatter(long h, torch::Tensor out, torch::Tensor in,
                    uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(out.is_cuda() && in.is_cuda());
  TORCH_CHECK(in.numel() == out.numel() * c.world,
              "reduce_scatter: input must be world_size * output");
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclReduceScatter(in.data_ptr(), out.data_ptr(),
                                 out.numel(), to_nccl_dtype(in), ncclSum,
                                 c.comm, s));
  });
}

long all_gather(long h, torch::Tensor out, torch::Tensor in,
                uintptr_t caller_stream) {
  auto& c = get(h);
  TORCH_CHECK(out.is_cuda() && in.is_cuda());
  TORCH_CHECK(out.numel() == in.numel() * c.world,
              "all_gather: output must be world_size * input");
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclAllGather(in.data_ptr(), out.data_ptr(), in.numel(),
                             to_nccl_dtype(in), c.comm, s));
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

// grouped all-reduce of several tensors in one RCCL group call
long all_reduce_coalesced(long h, std::vector<torch::Tensor> ts,
                          std::string op, uintptr_t caller_stream) {
  auto& c = get(h);
  return run_on_comm_stream(c, caller_stream, [&](hipStream_t s) {
    CHECK_NCCL(ncclGroupStart());
    for (auto& t : ts) {
      CHECK_NCCL(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                               to_nccl_dtype(t), to_nccl_op(op), c.comm,
                               s));
    }
    CHECK_NCCL(ncclGroupEnd());
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
}
