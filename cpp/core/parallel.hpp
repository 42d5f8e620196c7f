// grapehip — host-side thread parallelism.
//
// Reference parity: grape/utils/thread_pool.h + grape/parallel/parallel_engine.h
// (ForEach with atomic chunk-claiming work stealing). We keep a single
// process-wide pool of std::threads and a chunked parallel_for; apps use it
// for every hot loop on the CPU path.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstddef>
#include <functional>
#include <mutex>
#include <thread>
#include <vector>

namespace grapehip {

class ThreadPool {
 public:
  static ThreadPool& Get() {
    static ThreadPool pool;
    return pool;
  }

  int num_threads() const { return n_; }
  void set_num_threads(int n) {
    if (n > 0 && n != n_) {
      shutdown();
      start(n);
    }
  }

  // Run fn(tid) on all threads, block until done.
  void run(const std::function<void(int)>& fn) {
    std::unique_lock<std::mutex> lk(m_);
    fn_ = &fn;
    ++epoch_;
    pending_ = n_ - 1;
    cv_.notify_all();
    lk.unlock();
    fn(0);  // caller participates as tid 0
    lk.lock();
    done_cv_.wait(lk, [&] { return pending_ == 0; });
    fn_ = nullptr;
  }

  ~ThreadPool() { shutdown(); }

 private:
  ThreadPool() {
    unsigned hw = std::thread::hardware_concurrency();
    start(hw ? static_cast<int>(hw) : 4);
  }
  void start(int n) {
    n_ = n;
    stop_ = false;
    // workers start at the CURRENT epoch: a pool restarted after earlier
    // runs (set_num_threads) must not treat a stale epoch as a pending
    // task — a spurious wakeup would otherwise call a null fn_
    for (int t = 1; t < n_; ++t) {
      workers_.emplace_back([this, t, e = epoch_] {
        uint64_t seen = e;
        for (;;) {
          const std::function<void(int)>* fn;
          {
            std::unique_lock<std::mutex> lk(m_);
            cv_.wait(lk, [&] { return stop_ || epoch_ != seen; });
            if (stop_) return;
            seen = epoch_;
            fn = fn_;
          }
          (*fn)(t);
          {
            std::lock_guard<std::mutex> lk(m_);
            if (--pending_ == 0) done_cv_.notify_one();
          }
        }
      });
    }
  }
  void shutdown() {
    {
      std::lock_guard<std::mutex> lk(m_);
      stop_ = true;
      cv_.notify_all();
    }
    for (auto& w : workers_) w.join();
    workers_.clear();
  }

  int n_ = 0;
  bool stop_ = false;
  uint64_t epoch_ = 0;
  int pending_ = 0;
  const std::function<void(int)>* fn_ = nullptr;
  std::mutex m_;
  std::condition_variable cv_, done_cv_;
  std::vector<std::thread> workers_;
};

// Chunked work-stealing parallel for over [begin, end).
template <typename F>
inline void parallel_for(size_t begin, size_t end, F&& f,
                         size_t chunk = 1024) {
  if (end <= begin) return;
  size_t total = end - begin;
  auto& pool = ThreadPool::Get();
  if (total <= chunk || pool.num_threads() == 1) {
    for (size_t i = begin; i < end; ++i) f(i);
    return;
  }
  std::atomic<size_t> cursor{begin};
  pool.run([&](int) {
    for (;;) {
      size_t lo = cursor.fetch_add(chunk, std::memory_order_relaxed);
      if (lo >= end) break;
      size_t hi = lo + chunk < end ? lo + chunk : end;
      for (size_t i = lo; i < hi; ++i) f(i);
    }
  });
}

// Per-thread variant: f(tid, i). Useful for thread-local accumulators.
template <typename F>
inline void parallel_for_tid(size_t begin, size_t end, F&& f,
                             size_t chunk = 1024) {
  if (end <= begin) return;
  auto& pool = ThreadPool::Get();
  size_t total = end - begin;
  if (total <= chunk || pool.num_threads() == 1) {
    for (size_t i = begin; i < end; ++i) f(0, i);
    return;
  }
  std::atomic<size_t> cursor{begin};
  pool.run([&](int tid) {
    for (;;) {
      size_t lo = cursor.fetch_add(chunk, std::memory_order_relaxed);
      if (lo >= end) break;
      size_t hi = lo + chunk < end ? lo + chunk : end;
      for (size_t i = lo; i < hi; ++i) f(tid, i);
    }
  });
}

template <typename T>
inline bool atomic_min_update(std::atomic<T>& slot, T val) {
  T cur = slot.load(std::memory_order_relaxed);
  while (val < cur) {
    if (slot.compare_exchange_weak(cur, val, std::memory_order_relaxed))
      return true;
  }
  return false;
}

}  // namespace grapehip
