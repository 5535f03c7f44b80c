// Native RCCL communicator: a from-scratch replacement for
// the slice of ProcessGroupNCCL the reference's DDP path uses
// (/root/reference/cifar_example_ddp.py:57-58,83,106) — communicator
// bootstrap by ncclUniqueId exchange (TCP rendezvous done in Python,
// mi355x/parallel/rccl.py), and all_reduce / broadcast / all_gather /
// reduce_scatter / barrier over xGMI on a DEDICATED high-priority HIP
// stream so bucket all-reduces overlap the backward compute stream.
//
// Stream protocol per collective: record event on the producing (current
// torch) stream -> comm stream waits -> ncclX(...) on comm stream. wait()
// makes the current stream wait on the comm stream — no host sync.
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <stdexcept>
#include <string>

#define HIP_CHECK(x)                                                    \
  do {                                                                  \
    hipError_t e_ = (x);                                                \
    TORCH_CHECK(e_ == hipSuccess, "HIP error: ", hipGetErrorString(e_)); \
  } while (0)

#define NCCL_CHECK(x)                                                      \
  do {                                                                     \
    ncclResult_t r_ = (x);                                                 \
    TORCH_CHECK(r_ == ncclSuccess, "RCCL error: ", ncclGetErrorString(r_)); \
  } while (0)

namespace {

ncclDataType_t nccl_dtype(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat:
      return ncclFloat32;
    case at::kBFloat16:
      return ncclBfloat16;
    case at::kHalf:
      return ncclFloat16;
    case at::kDouble:
      return ncclFloat64;
    case at::kInt:
      return ncclInt32;
    case at::kLong:
      return ncclInt64;
    default:
      TORCH_CHECK(false, "unsupported dtype for RCCL collective");
  }
}

class RcclComm {
 public:
  RcclComm(int rank, int world, py::bytes unique_id) : rank_(rank), world_(world) {
    std::string id = unique_id;
    TORCH_CHECK(id.size() == sizeof(ncclUniqueId), "bad ncclUniqueId size");
    ncclUniqueId uid;
    memcpy(&uid, id.data(), sizeof(uid));
    int least, greatest;
    HIP_CHECK(hipDeviceGetStreamPriorityRange(&least, &greatest));
    HIP_CHECK(hipStreamCreateWithPriority(&stream_, hipStreamNonBlocking,
                                          greatest));
    HIP_CHECK(hipEventCreateWithFlags(&ev_in_, hipEventDisableTiming));
    HIP_CHECK(hipEventCreateWithFlags(&ev_out_, hipEventDisableTiming));
    NCCL_CHECK(ncclCommInitRank(&comm_, world, uid, rank));
  }

  ~RcclComm() {
    if (comm_) ncclCommDestroy(comm_);
    hipStreamDestroy(stream_);
    hipEventDestroy(ev_in_);
    hipEventDestroy(ev_out_);
  }

  int rank() const { return rank_; }
  int world_size() const { return world_; }

  void all_reduce(at::Tensor t) {
    check(t);
    sync_to_comm();
    NCCL_CHECK(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                             nccl_dtype(t), ncclSum, comm_, stream_));
  }

  void broadcast(at::Tensor t, int root) {
    check(t);
    sync_to_comm();
    NCCL_CHECK(ncclBroadcast(t.data_ptr(), t.data_ptr(), t.numel(),
                             nccl_dtype(t), root, comm_, stream_));
  }

  void all_gather(at::Tensor out, at::Tensor in) {
    check(in);
    TORCH_CHECK(out.numel() == in.numel() * world_, "all_gather size");
    sync_to_comm();
    NCCL_CHECK(ncclAllGather(in.data_ptr(), out.data_ptr(), in.numel(),
                             nccl_dtype(in), comm_, stream_));
  }

  void reduce_scatter(at::Tensor out, at::Tensor in) {
    check(out);
    TORCH_CHECK(in.numel() == out.numel() * world_, "reduce_scatter size");
    sync_to_comm();
    NCCL_CHECK(ncclReduceScatter(in.data_ptr(), out.data_ptr(), out.numel(),
                                 nccl_dtype(out), ncclSum, comm_, stream_));
  }

  // make the CURRENT torch stream wait for all queued collectives
  void wait() {
    HIP_CHECK(hipEventRecord(ev_out_, stream_));
    HIP_CHECK(hipStreamWaitEvent(
        (hipStream_t)at::cuda::getCurrentCUDAStream().stream(), ev_out_, 0));
  }

  void barrier() {
    if (!barrier_buf_.defined())
      barrier_buf_ = at::zeros({1}, at::TensorOptions()
                                        .dtype(at::kFloat)
                                        .device(at::kCUDA));
    all_reduce(barrier_buf_);
    HIP_CHECK(hipStreamSynchronize(stream_));
  }

 private:
  void check(const at::Tensor& t) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous(),
                "RCCL collective needs contiguous CUDA tensor");
  }
  void sync_to_comm() {
    HIP_CHECK(hipEventRecord(
        ev_in_, (hipStream_t)at::cuda::getCurrentCUDAStream().stream()));
    HIP_CHECK(hipStreamWaitEvent(stream_, ev_in_, 0));
  }

  int rank_, world_;
  ncclComm_t comm_ = nullptr;
  hipStream_t stream_{};
  hipEvent_t ev_in_{}, ev_out_{};
  at::Tensor barrier_buf_;
};

py::bytes get_unique_id() {
  ncclUniqueId uid;
  NCCL_CHECK(ncclGetUniqueId(&uid));
  return py::bytes(reinterpret_cast<const char*>(&uid), sizeof(uid));
}

}  // namespace

void register_rccl(py::module_& m) {
  m.def("rccl_get_unique_id", &get_unique_id);
  py::class_<RcclComm>(m, "RcclComm")
      .def(py::init<int, int, py::bytes>(), py::arg("rank"),
           py::arg("world_size"), py::arg("unique_id"))
      .def("rank", &RcclComm::rank)
      .def("world_size", &RcclComm::world_size)
      .def("all_reduce", &RcclComm::all_reduce)
      .def("broadcast", &RcclComm::broadcast, py::arg("tensor"),
           py::arg("root") = 0)
      .def("all_gather", &RcclComm::all_gather)
      .def("reduce_scatter", &RcclComm::reduce_scatter)
      .def("wait", &RcclComm::wait)
      .def("barrier", &RcclComm::barrier);
}
