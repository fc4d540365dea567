#include "comm/rccl_engine.h"

#include <functional>
#include <stdexcept>
#include <cstdio>
#include <fstream>

namespace ddl {

#define HIP_OK(x)                                                       \
  do {                                                                  \
    hipError_t e_ = (x);                                                \
    if (e_ != hipSuccess)                                               \
      throw std::runtime_error(std::string("HIP error: ") +            \
                               hipGetErrorString(e_) + " at " #x);     \
  } while (0)

#define NCCL_OK(x)                                                      \
  do {                                                                  \
    ncclResult_t r_ = (x);                                              \
    if (r_ != ncclSuccess)                                              \
      throw std::runtime_error(std::string("RCCL error: ") +           \
                               ncclGetErrorString(r_) + " at " #x);    \
  } while (0)

ncclDataType_t to_nccl_dtype(int code) {
  switch (code) {
    case 0: return ncclFloat32;
    case 1: return ncclBfloat16;
    case 2: return ncclFloat16;
    case 3: return ncclInt64;
    case 4: return ncclInt32;
    case 5: return ncclUint8;
    default: throw std::runtime_error("bad dtype code");
  }
}

static int64_t dtype_size(int code) {
  switch (code) {
    case 0: return 4; case 1: return 2; case 2: return 2;
    case 3: return 8; case 4: return 4; case 5: return 1;
    default: return 1;
  }
}

CommEngine::CommEngine(int rank, int size, int device,
                       const std::string& uid_bytes)
    : rank_(rank), size_(size), device_(device) {
  if (uid_bytes.size() != sizeof(ncclUniqueId))
    throw std::runtime_error("bad ncclUniqueId payload");
  HIP_OK(hipSetDevice(device_));
  HIP_OK(hipStreamCreateWithFlags(&comm_stream_, hipStreamNonBlocking));
  ncclUniqueId uid;
  memcpy(&uid, uid_bytes.data(), sizeof(uid));
  NCCL_OK(ncclCommInitRank(&comm_, size_, uid, rank_));
  t0_ = std::chrono::steady_clock::now();
}

CommEngine::~CommEngine() { shutdown(); }

void CommEngine::shutdown() {
  std::lock_guard<std::mutex> lk(mu_);
  if (comm_) {
    ncclCommDestroy(comm_);
    comm_ = nullptr;
  }
  if (comm_stream_) {
    hipStreamDestroy(comm_stream_);
    comm_stream_ = nullptr;
  }
  for (auto& kv : inflight_) hipEventDestroy(kv.second.done);
  inflight_.clear();
  while (!event_pool_.empty()) {
    hipEventDestroy(event_pool_.front());
    event_pool_.pop_front();
  }
  if (timeline_on_) dump_timeline();
}

hipEvent_t CommEngine::get_event() {
  if (!event_pool_.empty()) {
    hipEvent_t e = event_pool_.front();
    event_pool_.pop_front();
    return e;
  }
  hipEvent_t e;
  HIP_OK(hipEventCreateWithFlags(&e, hipEventDisableTiming));
  return e;
}

void CommEngine::put_event(hipEvent_t e) { event_pool_.push_back(e); }

int64_t CommEngine::finish_enqueue(hipStream_t ready_stream, const char* what,
                                   int64_t bytes,
                                   const std::function<void()>& collective) {
  std::lock_guard<std::mutex> lk(mu_);
  // compute -> comm handshake
  hipEvent_t ready = get_event();
  HIP_OK(hipEventRecord(ready, ready_stream));
  HIP_OK(hipStreamWaitEvent(comm_stream_, ready, 0));
  put_event(ready);
  auto t_start = std::chrono::steady_clock::now();
  collective();
  hipEvent_t done = get_event();
  HIP_OK(hipEventRecord(done, comm_stream_));
  int64_t h = next_handle_++;
  char label[96];
  snprintf(label, sizeof(label), "%s#%lld/%lld MB", what,
           (long long)h, (long long)(bytes >> 20));
  inflight_[h] = Inflight{done, t_start, label};
  if (timeline_on_) {
    double ts = std::chrono::duration<double, std::micro>(t_start - t0_).count();
    double te = std::chrono::duration<double, std::micro>(
                    std::chrono::steady_clock::now() - t0_).count();
    timeline_.push_back({std::string("enqueue/") + label, ts, te - ts});
  }
  return h;
}

int64_t CommEngine::allreduce_async(void* data, int64_t numel, int dtype_code,
                                    bool average, hipStream_t ready_stream) {
  auto dt = to_nccl_dtype(dtype_code);
  ncclRedOp_t op = average && size_ > 1 ? ncclAvg : ncclSum;
  return finish_enqueue(ready_stream, "allreduce",
                        numel * dtype_size(dtype_code), [&] {
    NCCL_OK(ncclAllReduce(data, data, numel, dt, op, comm_, comm_stream_));
  });
}

int64_t CommEngine::broadcast_async(void* data, int64_t numel, int dtype_code,
                                    int root, hipStream_t ready_stream) {
  auto dt = to_nccl_dtype(dtype_code);
  return finish_enqueue(ready_stream, "broadcast",
                        numel * dtype_size(dtype_code), [&] {
    NCCL_OK(ncclBroadcast(data, data, numel, dt, root, comm_, comm_stream_));
  });
}

int64_t CommEngine::allgather_async(const void* src, void* dst,
                                    int64_t send_numel, int dtype_code,
                                    hipStream_t ready_stream) {
  auto dt = to_nccl_dtype(dtype_code);
  return finish_enqueue(ready_stream, "allgather",
                        send_numel * dtype_size(dtype_code) * size_, [&] {
    NCCL_OK(ncclAllGather(src, dst, send_numel, dt, comm_, comm_stream_));
  });
}

void CommEngine::wait_handle(int64_t handle, hipStream_t stream) {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = inflight_.find(handle);
  if (it == inflight_.end()) return;  // already waited
  HIP_OK(hipStreamWaitEvent(stream, it->second.done, 0));
  put_event(it->second.done);
  if (timeline_on_) {
    double ts = std::chrono::duration<double, std::micro>(
                    it->second.enqueued - t0_).count();
    double te = std::chrono::duration<double, std::micro>(
                    std::chrono::steady_clock::now() - t0_).count();
    timeline_.push_back({std::string("span/") + it->second.label, ts, te - ts});
  }
  inflight_.erase(it);
}

void CommEngine::join_stream(hipStream_t stream) {
  std::vector<int64_t> hs;
  {
    std::lock_guard<std::mutex> lk(mu_);
    for (auto& kv : inflight_) hs.push_back(kv.first);
  }
  for (int64_t h : hs) wait_handle(h, stream);
}

bool CommEngine::is_complete(int64_t handle) {
  std::lock_guard<std::mutex> lk(mu_);
  auto it = inflight_.find(handle);
  if (it == inflight_.end()) return true;
  return hipEventQuery(it->second.done) == hipSuccess;
}

int64_t CommEngine::outstanding() const {
  std::lock_guard<std::mutex> lk(mu_);
  return (int64_t)inflight_.size();
}

double CommEngine::oldest_age_seconds() const {
  std::lock_guard<std::mutex> lk(mu_);
  double worst = 0.0;
  auto now = std::chrono::steady_clock::now();
  for (auto& kv : inflight_) {
    double age =
        std::chrono::duration<double>(now - kv.second.enqueued).count();
    if (age > worst) worst = age;
  }
  return worst;
}

void CommEngine::enable_timeline(const std::string& path) {
  timeline_on_ = true;
  timeline_path_ = path;
}

void CommEngine::dump_timeline() {
  if (!timeline_on_ || timeline_path_.empty()) return;
  std::ofstream f(timeline_path_);
  f << "{\"traceEvents\":[";
  for (size_t i = 0; i < timeline_.size(); ++i) {
    const auto& e = timeline_[i];
    if (i) f << ",";
    f << "{\"name\":\"" << e.name << "\",\"ph\":\"X\",\"ts\":" << e.ts_us
      << ",\"dur\":" << e.dur_us << ",\"pid\":" << rank_ << ",\"tid\":1}";
  }
  f << "]}";
}

}  // namespace ddl
