// Native RCCL comm engine — the MI355X replacement for Horovod's C++ core
// (horovod/common/operations.cc + ops/nccl_operations.cc).  Differences by
// design: no background thread and no negotiation round —
// bucket order is static, so every collective is enqueued synchronously from
// the autograd hook onto a dedicated side HIP stream, with hipEvent
// handshakes both ways (compute→comm before the collective, comm→compute at
// synchronize).  The host never blocks; overlap with backward is pure
// stream-level concurrency over xGMI.
#pragma once
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <chrono>
#include <cstdint>
#include <deque>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

namespace ddl {

struct TimelineEvent {
  std::string name;
  double ts_us;
  double dur_us;
};

class CommEngine {
 public:
  CommEngine(int rank, int size, int device, const std::string& uid_bytes);
  ~CommEngine();

  int rank() const { return rank_; }
  int size() const { return size_; }

  // Enqueue on the side stream; returns a handle id.  `ready_stream` is the
  // producer (compute) stream the collective must wait on.
  int64_t allreduce_async(void* data, int64_t numel, int dtype_code,
                          bool average, hipStream_t ready_stream);
  int64_t broadcast_async(void* data, int64_t numel, int dtype_code, int root,
                          hipStream_t ready_stream);
  int64_t allgather_async(const void* src, void* dst, int64_t send_numel,
                          int dtype_code, hipStream_t ready_stream);

  // Make `stream` (the compute stream) wait for handle completion.
  // Device-side only.
  void wait_handle(int64_t handle, hipStream_t stream);
  // Wait every outstanding handle (device-side) on `stream`.
  void join_stream(hipStream_t stream);
  // Host-side poll: true if the collective finished on the GPU.
  bool is_complete(int64_t handle);

  int64_t outstanding() const;
  double oldest_age_seconds() const;

  // Chrome-trace timeline (host enqueue spans).
  void enable_timeline(const std::string& path);
  void dump_timeline();

  void shutdown();

 private:
  struct Inflight {
    hipEvent_t done;
    std::chrono::steady_clock::time_point enqueued;
    std::string label;
  };
  hipEvent_t get_event();
  void put_event(hipEvent_t e);
  int64_t finish_enqueue(hipStream_t ready_stream, const char* what,
                         int64_t bytes,
                         const std::function<void()>& collective);

  int rank_ = 0, size_ = 1, device_ = 0;
  ncclComm_t comm_ = nullptr;
  hipStream_t comm_stream_ = nullptr;
  std::deque<hipEvent_t> event_pool_;
  std::unordered_map<int64_t, Inflight> inflight_;
  int64_t next_handle_ = 1;
  mutable std::mutex mu_;
  bool timeline_on_ = false;
  std::string timeline_path_;
  std::vector<TimelineEvent> timeline_;
  std::chrono::steady_clock::time_point t0_;
};

ncclDataType_t to_nccl_dtype(int code);

}  // namespace ddl
