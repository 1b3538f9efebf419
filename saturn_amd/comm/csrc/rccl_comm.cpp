// Native RCCL communicator for saturn_amd (SURVEY C1-C7 worklist).
//
// Hand-written C++ on RCCL-over-xGMI: one communicator per gang-scheduled
// job (created/destroyed per interval — the gang lifecycle of SURVEY §7
// hard-part 3), with a dedicated high-priority HIP stream and event-based
// compute<->comm ordering so bucketed all-reduces overlap backward
// (DDP data plane), plus all-gather / reduce-scatter / broadcast for the
// ZeRO-3 shard manager.
//
// The reference reached all of this through torch DDP/FSDP internals
// (examples/wikitext103/executors/DDP.py:90, FSDP.py:184); this module is
// the framework's own comm layer.  Bootstrap: rank 0 creates the
// ncclUniqueId, peers receive it out of band (the Python wrapper ships it
// through the job's 127.0.0.1 rendezvous, port pool keyed by task id).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <string>
#include <vector>

namespace samd_comm {

#define RCCL_CHECK(cmd)                                                   \
  do {                                                                    \
    ncclResult_t r = (cmd);                                               \
    TORCH_CHECK(r == ncclSuccess, "RCCL error: ", ncclGetErrorString(r)); \
  } while (0)

#define HIP_CHECK(cmd)                                                     \
  do {                                                                     \
    hipError_t e = (cmd);                                                  \
    TORCH_CHECK(e == hipSuccess, "HIP error: ", hipGetErrorString(e));     \
  } while (0)

static ncclDataType_t dtype_of(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kBFloat16: return ncclBfloat16;
    case at::kHalf: return ncclHalf;
    case at::kFloat: return ncclFloat;
    case at::kDouble: return ncclDouble;
    case at::kInt: return ncclInt32;
    case at::kLong: return ncclInt64;
    default: TORCH_CHECK(false, "unsupported dtype for RCCL");
  }
}

class RcclComm {
 public:
  RcclComm(py::bytes unique_id, int rank, int world) : rank_(rank), world_(world) {
    std::string id_str = unique_id;
    TORCH_CHECK(id_str.size() == sizeof(ncclUniqueId), "bad unique id size");
    ncclUniqueId id;
    memcpy(&id, id_str.data(), sizeof(id));
    int least = 0, greatest = 0;
    HIP_CHECK(hipDeviceGetStreamPriorityRange(&least, &greatest));
    HIP_CHECK(hipStreamCreateWithPriority(&stream_, hipStreamNonBlocking,
                                          greatest));
    HIP_CHECK(hipEventCreateWithFlags(&entry_ev_, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&done_ev_, hipEventDisableTiming));
    RCCL_CHECK(ncclCommInitRank(&comm_, world, id, rank));
  }

  ~RcclComm() {
    // never tear down under in-flight collectives (gang churn destroys a
    // communicator right after an interval's last bucket)
    if (stream_) (void)hipStreamSynchronize(stream_);
    if (comm_) ncclCommDestroy(comm_);
    if (stream_) hipStreamDestroy(stream_);
    if (entry_ev_) hipEventDestroy(entry_ev_);
    if (done_ev_) hipEventDestroy(done_ev_);
  }

  // Make the comm stream wait for work queued on torch's current stream.
  void fence_compute() {
    auto cur = at::hip::getCurrentHIPStream().stream();
    HIP_CHECK(hipEventRecord(entry_ev_, cur));
    HIP_CHECK(hipStreamWaitEvent(stream_, entry_ev_, 0));
  }

  void all_reduce(at::Tensor t, bool average) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    fence_compute();
    RCCL_CHECK(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                             dtype_of(t), average ? ncclAvg : ncclSum, comm_,
                             stream_));
  }

  void broadcast(at::Tensor t, int root) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    fence_compute();
    RCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                             dtype_of(t), root, comm_, stream_));
  }

  void all_gather(at::Tensor out, at::Tensor in) {
    TORCH_CHECK(out.is_cuda() && in.is_cuda());
    TORCH_CHECK(out.numel() == in.numel() * world_);
    fence_compute();
    RCCL_CHECK(ncclAllGather(in.data_ptr(), out.data_ptr(), in.numel(),
                             dtype_of(in), comm_, stream_));
  }

  void reduce_scatter(at::Tensor out, at::Tensor in, bool average) {
    TORCH_CHECK(out.is_cuda() && in.is_cuda());
    TORCH_CHECK(in.numel() == out.numel() * world_);
    fence_compute();
    RCCL_CHECK(ncclReduceScatter(in.data_ptr(), out.data_ptr(), out.numel(),
                                 dtype_of(in), average ? ncclAvg : ncclSum,
                                 comm_, stream_));
  }

  void send(at::Tensor t, int peer) {
    fence_compute();
    RCCL_CHECK(ncclSend(t.data_ptr(), t.numel(), dtype_of(t), peer, comm_,
                        stream_));
  }

  void recv(at::Tensor t, int peer) {
    fence_compute();
    RCCL_CHECK(ncclRecv(t.data_ptr(), t.numel(), dtype_of(t), peer, comm_,
                        stream_));
  }

  // Queue-ordered barrier: torch's current stream waits for everything
  // enqueued on the comm stream so far.
  void join() {
    HIP_CHECK(hipEventRecord(done_ev_, stream_));
    auto cur = at::hip::getCurrentHIPStream().stream();
    HIP_CHECK(hipStreamWaitEvent(cur, done_ev_, 0));
  }

  void synchronize() { HIP_CHECK(hipStreamSynchronize(stream_)); }

  int rank() const { return rank_; }
  int world() const { return world_; }

 private:
  ncclComm_t comm_ = nullptr;
  hipStream_t stream_ = nullptr;
  hipEvent_t entry_ev_ = nullptr;
  hipEvent_t done_ev_ = nullptr;
  int rank_;
  int world_;
};

static py::bytes get_unique_id() {
  ncclUniqueId id;
  RCCL_CHECK(ncclGetUniqueId(&id));
  return py::bytes(reinterpret_cast<const char*>(&id), sizeof(id));
}

}  // namespace samd_comm

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "saturn_amd native RCCL comm engine";
  m.def("get_unique_id", &samd_comm::get_unique_id);
  py::class_<samd_comm::RcclComm>(m, "RcclComm")
      .def(py::init<py::bytes, int, int>(), py::arg("unique_id"),
           py::arg("rank"), py::arg("world"))
      .def("all_reduce", &samd_comm::RcclComm::all_reduce, py::arg("tensor"),
           py::arg("average") = true)
      .def("broadcast", &samd_comm::RcclComm::broadcast)
      .def("all_gather", &samd_comm::RcclComm::all_gather)
      .def("reduce_scatter", &samd_comm::RcclComm::reduce_scatter,
           py::arg("out"), py::arg("in"), py::arg("average") = false)
      .def("send", &samd_comm::RcclComm::send)
      .def("recv", &samd_comm::RcclComm::recv)
      .def("join", &samd_comm::RcclComm::join)
      .def("synchronize", &samd_comm::RcclComm::synchronize)
      .def_property_readonly("rank", &samd_comm::RcclComm::rank)
      .def_property_readonly("world", &samd_comm::RcclComm::world);
}
