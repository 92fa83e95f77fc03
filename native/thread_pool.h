// Minimal fixed-size thread pool for stripe-parallel CPU encode.
//
// Latency-tuned: keep pools SMALL (the encode pipelines cap at 16) —
// waking a hardware_concurrency-sized pool costs more in cv wakeups
// than the per-frame work. Workers can optionally micro-spin on the
// job counter before sleeping (HIPFLUX_POOL_SPIN_US, default 0):
// measured neutral for one session and -25% aggregate with 8 sessions
// per GPU (8 x 16 spinning threads), so it stays off by default.
#pragma once

#include <atomic>
#include <chrono>
#include <condition_variable>
#include <cstdlib>
#include <functional>
#include <mutex>
#include <queue>
#include <thread>
#include <vector>

namespace hipflux {

class ThreadPool {
 public:
  explicit ThreadPool(int n) {
    int spin_us = 0;
    if (const char* e = std::getenv("HIPFLUX_POOL_SPIN_US"))
      spin_us = std::atoi(e);
    for (int i = 0; i < n; ++i)
      workers_.emplace_back([this, spin_us] {
        for (;;) {
          std::function<void()> job;
          {
            std::unique_lock<std::mutex> lk(mu_);
            if (stop_) {
              if (jobs_.empty()) return;
            } else if (jobs_.empty() && spin_us > 0) {
              // micro-spin before sleeping: drop the lock and poll
              lk.unlock();
              auto until = std::chrono::steady_clock::now() +
                           std::chrono::microseconds(spin_us);
              while (pending_.load(std::memory_order_relaxed) == 0 &&
                     !stop_flag_.load(std::memory_order_relaxed) &&
                     std::chrono::steady_clock::now() < until)
                ;
              lk.lock();
            }
            cv_.wait(lk, [this] { return stop_ || !jobs_.empty(); });
            if (stop_ && jobs_.empty()) return;
            job = std::move(jobs_.front());
            jobs_.pop();
            pending_.fetch_sub(1, std::memory_order_relaxed);
          }
          job();
          {
            std::lock_guard<std::mutex> lk(mu_);
            if (--outstanding_ == 0) done_cv_.notify_all();
          }
        }
      });
  }

  ~ThreadPool() {
    {
      std::lock_guard<std::mutex> lk(mu_);
      stop_ = true;
      stop_flag_.store(true, std::memory_order_relaxed);
    }
    cv_.notify_all();
    for (auto& w : workers_) w.join();
  }

  void submit(std::function<void()> job) {
    {
      std::lock_guard<std::mutex> lk(mu_);
      ++outstanding_;
      jobs_.push(std::move(job));
      pending_.fetch_add(1, std::memory_order_relaxed);
    }
    cv_.notify_one();
  }

  void wait_all() {
    std::unique_lock<std::mutex> lk(mu_);
    done_cv_.wait(lk, [this] { return outstanding_ == 0; });
  }

  int size() const { return static_cast<int>(workers_.size()); }

 private:
  std::vector<std::thread> workers_;
  std::queue<std::function<void()>> jobs_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::condition_variable done_cv_;
  int outstanding_ = 0;
  bool stop_ = false;
  std::atomic<int> pending_{0};
  std::atomic<bool> stop_flag_{false};
};

}  // namespace hipflux
