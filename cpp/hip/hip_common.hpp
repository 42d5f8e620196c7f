// grapehip — HIP substrate: error checks, buffers, streams, wave utils.
// Reference parity: grape/cuda/utils/{stream,event,device_buffer,
// shared_value,array_view}.h — rebuilt for HIP/CDNA4 (wave64, gfx950).
#pragma once

#include <hip/hip_runtime.h>

#include <atomic>
#include <cstdint>
#include <stdexcept>
#include <string>
#include <vector>

// rocTX range markers (reference NVTX RangeMarker, cuda/utils/markers.h):
// enabled with GRAPEHIP_ROCTX=1; the library is dlopen'd so builds carry
// no hard dependency. View ranges with `rocprofv3 --marker-trace`.
#include <dlfcn.h>

namespace grapehip {
namespace roctx {
using PushFn = int (*)(const char*);
using PopFn = int (*)();
inline PushFn& push_fn() {
  static PushFn fn = [] {
    if (!getenv("GRAPEHIP_ROCTX")) return static_cast<PushFn>(nullptr);
    void* h = dlopen("libroctx64.so", RTLD_NOW | RTLD_GLOBAL);
    return h ? reinterpret_cast<PushFn>(dlsym(h, "roctxRangePushA"))
             : nullptr;
  }();
  return fn;
}
inline PopFn& pop_fn() {
  static PopFn fn = [] {
    if (!getenv("GRAPEHIP_ROCTX")) return static_cast<PopFn>(nullptr);
    void* h = dlopen("libroctx64.so", RTLD_NOW | RTLD_GLOBAL);
    return h ? reinterpret_cast<PopFn>(dlsym(h, "roctxRangePop")) : nullptr;
  }();
  return fn;
}
}  // namespace roctx

struct RangeMarker {
  explicit RangeMarker(const char* name) {
    if (roctx::push_fn()) roctx::push_fn()(name);
  }
  ~RangeMarker() {
    if (roctx::pop_fn()) roctx::pop_fn()();
  }
};
}  // namespace grapehip

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error at ") + __FILE__ +    \
                               ":" + std::to_string(__LINE__) + ": " +      \
                               hipGetErrorString(_e));                      \
    }                                                                       \
  } while (0)

namespace grapehip {

constexpr int kWave = 64;          // CDNA wavefront width
constexpr int kBlock = 256;        // default workgroup (4 waves)
// Memory-bound grid cap: 256 CUs x 8 blocks (cdna guide G11), grid-stride
// the remainder.
constexpr int kMaxGrid = 2048;

inline int grid_for(size_t n, int block = kBlock, int cap = kMaxGrid) {
  if (n == 0) return 1;
  size_t g = (n + block - 1) / block;
  return static_cast<int>(g < static_cast<size_t>(cap) ? g : cap);
}

// HBM allocation tracking (reference utils/memory_tracker.{h,cc},
// TRACKING_MEMORY): every DeviceBuffer counts toward current + peak,
// surfaced by Engine.memory_info().
inline std::atomic<long long>& hip_alloc_current() {
  static std::atomic<long long> v{0};
  return v;
}
inline std::atomic<long long>& hip_alloc_peak() {
  static std::atomic<long long> v{0};
  return v;
}
inline void hip_alloc_note(long long delta) {
  long long cur = hip_alloc_current().fetch_add(delta) + delta;
  if (delta > 0) {
    long long p = hip_alloc_peak().load();
    while (cur > p && !hip_alloc_peak().compare_exchange_weak(p, cur)) {
    }
  }
}

// ---------------------------------------------------------------------------
template <typename T>
class DeviceBuffer {
 public:
  DeviceBuffer() = default;
  explicit DeviceBuffer(size_t n) { resize(n); }
  ~DeviceBuffer() { free(); }
  DeviceBuffer(const DeviceBuffer&) = delete;
  DeviceBuffer& operator=(const DeviceBuffer&) = delete;
  DeviceBuffer(DeviceBuffer&& o) noexcept : p_(o.p_), n_(o.n_) {
    o.p_ = nullptr;
    o.n_ = 0;
  }
  DeviceBuffer& operator=(DeviceBuffer&& o) noexcept {
    free();
    p_ = o.p_;
    n_ = o.n_;
    o.p_ = nullptr;
    o.n_ = 0;
    return *this;
  }

  void resize(size_t n) {
    if (n == n_) return;
    free();
    if (n) {
      HIP_CHECK(hipMalloc(&p_, n * sizeof(T)));
      hip_alloc_note(static_cast<long long>(n * sizeof(T)));
    }
    n_ = n;
  }
  void free() {
    if (p_) {
      (void)hipFree(p_);
      hip_alloc_note(-static_cast<long long>(n_ * sizeof(T)));
    }
    p_ = nullptr;
    n_ = 0;
  }
  T* data() { return p_; }
  const T* data() const { return p_; }
  size_t size() const { return n_; }

  void upload(const T* host, size_t n, hipStream_t s = nullptr) {
    resize(n);
    HIP_CHECK(hipMemcpyAsync(p_, host, n * sizeof(T), hipMemcpyHostToDevice, s));
  }
  void upload(const std::vector<T>& v, hipStream_t s = nullptr) {
    upload(v.data(), v.size(), s);
  }
  std::vector<T> download(hipStream_t s = nullptr) const {
    std::vector<T> out(n_);
    HIP_CHECK(hipMemcpyAsync(out.data(), p_, n_ * sizeof(T),
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    return out;
  }
  void zero(hipStream_t s = nullptr) {
    if (n_) HIP_CHECK(hipMemsetAsync(p_, 0, n_ * sizeof(T), s));
  }
  void fill_bytes(int byte, hipStream_t s = nullptr) {
    if (n_) HIP_CHECK(hipMemsetAsync(p_, byte, n_ * sizeof(T), s));
  }

 private:
  T* p_ = nullptr;
  size_t n_ = 0;
};

// Single device counter mirrored in pinned host memory.
class DeviceCounter {
 public:
  DeviceCounter() {
    HIP_CHECK(hipMalloc(&d_, sizeof(uint64_t)));
    HIP_CHECK(hipHostMalloc(&h_, sizeof(uint64_t)));
  }
  ~DeviceCounter() {
    if (d_) (void)hipFree(d_);
    if (h_) (void)hipHostFree(h_);
  }
  DeviceCounter(const DeviceCounter&) = delete;
  DeviceCounter& operator=(const DeviceCounter&) = delete;

  uint64_t* dev() { return d_; }
  void set(uint64_t v, hipStream_t s) {
    *h_ = v;
    HIP_CHECK(hipMemcpyAsync(d_, h_, 8, hipMemcpyHostToDevice, s));
  }
  uint64_t get(hipStream_t s) {
    HIP_CHECK(hipMemcpyAsync(h_, d_, 8, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    return *h_;
  }

 private:
  uint64_t* d_ = nullptr;
  uint64_t* h_ = nullptr;
};

class Stream {
 public:
  Stream() { HIP_CHECK(hipStreamCreateWithFlags(&s_, hipStreamNonBlocking)); }
  ~Stream() {
    if (s_) (void)hipStreamDestroy(s_);
  }
  Stream(const Stream&) = delete;
  Stream& operator=(const Stream&) = delete;
  hipStream_t get() const { return s_; }
  operator hipStream_t() const { return s_; }
  void sync() const { HIP_CHECK(hipStreamSynchronize(s_)); }

 private:
  hipStream_t s_ = nullptr;
};

class Event {
 public:
  Event() { HIP_CHECK(hipEventCreateWithFlags(&e_, hipEventDisableTiming)); }
  ~Event() {
    if (e_) (void)hipEventDestroy(e_);
  }
  Event(const Event&) = delete;
  Event& operator=(const Event&) = delete;
  void record(hipStream_t s) { HIP_CHECK(hipEventRecord(e_, s)); }
  void wait_on(hipStream_t s) { HIP_CHECK(hipStreamWaitEvent(s, e_, 0)); }
  void sync() { HIP_CHECK(hipEventSynchronize(e_)); }

 private:
  hipEvent_t e_ = nullptr;
};

}  // namespace grapehip
