// pybind11 surface of the native RCCL comm engine (_C_comm).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "comm/rccl_engine.h"

namespace {

int comm_dt(const at::Tensor& t) {
  switch (t.scalar_type()) {
    case at::kFloat: return 0;
    case at::kBFloat16: return 1;
    case at::kHalf: return 2;
    case at::kLong: return 3;
    case at::kInt: return 4;
    case at::kByte: return 5;
    default: TORCH_CHECK(false, "unsupported comm dtype ", t.scalar_type());
  }
}

hipStream_t cur_stream() { return c10::hip::getCurrentHIPStream().stream(); }

py::bytes get_unique_id() {
  ncclUniqueId uid;
  auto r = ncclGetUniqueId(&uid);
  TORCH_CHECK(r == ncclSuccess, "ncclGetUniqueId failed");
  return py::bytes(reinterpret_cast<const char*>(&uid), sizeof(uid));
}

class PyCommEngine {
 public:
  PyCommEngine(int rank, int size, int device, py::bytes uid)
      : eng_(rank, size, device, std::string(uid)) {}

  int rank() const { return eng_.rank(); }
  int size() const { return eng_.size(); }

  int64_t allreduce_async(at::Tensor t, bool average) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    return eng_.allreduce_async(t.data_ptr(), t.numel(), comm_dt(t), average,
                                cur_stream());
  }
  int64_t broadcast_async(at::Tensor t, int root) {
    TORCH_CHECK(t.is_cuda() && t.is_contiguous());
    return eng_.broadcast_async(t.data_ptr(), t.numel(), comm_dt(t), root,
                                cur_stream());
  }
  int64_t allgather_async(at::Tensor src, at::Tensor dst) {
    TORCH_CHECK(src.is_cuda() && src.is_contiguous() && dst.is_contiguous());
    return eng_.allgather_async(src.data_ptr(), dst.data_ptr(), src.numel(),
                                comm_dt(src), cur_stream());
  }
  void wait_handle(int64_t h) { eng_.wait_handle(h, cur_stream()); }
  void join_compute_stream() { eng_.join_stream(cur_stream()); }
  bool is_complete(int64_t h) { return eng_.is_complete(h); }
  int64_t outstanding() { return eng_.outstanding(); }
  double oldest_age_seconds() { return eng_.oldest_age_seconds(); }
  void enable_timeline(const std::string& p) { eng_.enable_timeline(p); }
  void dump_timeline() { eng_.dump_timeline(); }
  void shutdown() { eng_.shutdown(); }

 private:
  ddl::CommEngine eng_;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("get_unique_id", &get_unique_id);
  py::class_<PyCommEngine>(m, "CommEngine")
      .def(py::init<int, int, int, py::bytes>())
      .def("rank", &PyCommEngine::rank)
      .def("size", &PyCommEngine::size)
      .def("allreduce_async", &PyCommEngine::allreduce_async)
      .def("broadcast_async", &PyCommEngine::broadcast_async)
      .def("allgather_async", &PyCommEngine::allgather_async)
      .def("wait_handle", &PyCommEngine::wait_handle)
      .def("join_compute_stream", &PyCommEngine::join_compute_stream)
      .def("is_complete", &PyCommEngine::is_complete)
      .def("outstanding", &PyCommEngine::outstanding)
      .def("oldest_age_seconds", &PyCommEngine::oldest_age_seconds)
      .def("enable_timeline", &PyCommEngine::enable_timeline)
      .def("dump_timeline", &PyCommEngine::dump_timeline)
      .def("shutdown", &PyCommEngine::shutdown);
}
