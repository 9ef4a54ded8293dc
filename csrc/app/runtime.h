// Native runtime utilities: leveled logger, crash handler, SPSC work queue,
// thread affinity.  Capability parity with the reference's
// log/log.hpp:23-128, util/termination_handler.hpp:38-113, work.hpp:30-72
// (boost::lockfree::spsc_queue cap 2) and util/thread_affinity.hpp:34-122
// (hwloc) — reimplemented dependency-free for this framework.
#pragma once

#include <atomic>
#include <array>
#include <chrono>
#include <csignal>
#include <cstdlib>
#include <cstdio>
#include <cstring>
#include <execinfo.h>
#include <mutex>
#include <pthread.h>
#include <sched.h>
#include <sstream>
#include <string>
#include <condition_variable>
#include <deque>
#include <functional>
#include <vector>
#include <thread>
#include <unistd.h>

namespace srtb_app {

// ---------------- logging ----------------

enum class LogLevel : int { kNone = 0, kError = 1, kWarning = 2, kInfo = 3,
                            kDebug = 4 };

inline std::atomic<int>& log_level() {
  // default from SRTB_LOG_LEVEL env like the reference (log/log.hpp)
  static std::atomic<int> level{[] {
    const char* e = std::getenv("SRTB_LOG_LEVEL");
    return e ? std::atoi(e) : 3;
  }()};
  return level;
}

inline double log_uptime() {
  using clock = std::chrono::steady_clock;
  static const clock::time_point t0 = clock::now();
  return std::chrono::duration<double>(clock::now() - t0).count();
}

inline void log_write(LogLevel lv, const std::string& msg) {
  if ((int)lv > log_level().load(std::memory_order_relaxed)) return;
  static std::mutex mu;
  const char* tag = "";
  const char* color = "";
  switch (lv) {
    case LogLevel::kError: tag = "E"; color = "\033[31m"; break;
    case LogLevel::kWarning: tag = "W"; color = "\033[33m"; break;
    case LogLevel::kInfo: tag = "I"; color = "\033[32m"; break;
    case LogLevel::kDebug: tag = "D"; color = "\033[36m"; break;
    default: break;
  }
  std::lock_guard<std::mutex> lk(mu);
  std::fprintf(stderr, "%s[%s %10.3f]\033[0m %s\n", color, tag, log_uptime(),
               msg.c_str());
}

#define SRTB_APP_LOG(lv, expr)                        \
  do {                                                \
    std::ostringstream os_;                           \
    os_ << expr;                                      \
    ::srtb_app::log_write(lv, os_.str());             \
  } while (0)
#define SRTB_APP_LOGE(expr) SRTB_APP_LOG(::srtb_app::LogLevel::kError, expr)
#define SRTB_APP_LOGW(expr) SRTB_APP_LOG(::srtb_app::LogLevel::kWarning, expr)
#define SRTB_APP_LOGI(expr) SRTB_APP_LOG(::srtb_app::LogLevel::kInfo, expr)
#define SRTB_APP_LOGD(expr) SRTB_APP_LOG(::srtb_app::LogLevel::kDebug, expr)

// ---------------- termination / crash handler ----------------
// Prints a backtrace on fatal signals then chains to the default action
// (reference termination_handler.hpp uses boost::stacktrace).

inline void print_backtrace() {
  void* frames[64];
  const int n = backtrace(frames, 64);
  backtrace_symbols_fd(frames, n, STDERR_FILENO);
}

inline void crash_handler(int sig) {
  std::fprintf(stderr, "\n[srtb] fatal signal %d (%s); backtrace:\n", sig,
               strsignal(sig));
  print_backtrace();
  std::signal(sig, SIG_DFL);
  std::raise(sig);
}

inline void install_termination_handler() {
  for (int sig : {SIGSEGV, SIGILL, SIGABRT, SIGFPE, SIGBUS}) {
    std::signal(sig, crash_handler);
  }
  std::set_terminate([] {
    std::fprintf(stderr, "\n[srtb] std::terminate; backtrace:\n");
    print_backtrace();
    std::abort();
  });
}

// ---------------- SPSC bounded queue ----------------
// Lock-free single-producer/single-consumer ring, default capacity 2 like
// the reference's work queues (bounded depth = back-pressure).

template <typename T, size_t CAP = 2>
class SpscQueue {
 public:
  using value_type = T;
  bool try_push(T v) {
    const size_t w = w_.load(std::memory_order_relaxed);
    const size_t r = r_.load(std::memory_order_acquire);
    if (w - r >= CAP) return false;
    buf_[w % CAP] = std::move(v);
    w_.store(w + 1, std::memory_order_release);
    return true;
  }

  bool try_pop(T& out) {
    const size_t r = r_.load(std::memory_order_relaxed);
    const size_t w = w_.load(std::memory_order_acquire);
    if (r == w) return false;
    out = std::move(buf_[r % CAP]);
    r_.store(r + 1, std::memory_order_release);
    return true;
  }

  // blocking helpers with the configured wait (reference pipe_io.hpp:44-75)
  template <typename Stop>
  bool push(T v, Stop stop, size_t wait_ns = 1000) {
    while (!try_push(std::move(v))) {
      if (stop()) return false;
      std::this_thread::sleep_for(std::chrono::nanoseconds(wait_ns));
    }
    return true;
  }

  template <typename Stop>
  bool pop(T& out, Stop stop, size_t wait_ns = 1000) {
    while (!try_pop(out)) {
      if (stop()) return false;
      std::this_thread::sleep_for(std::chrono::nanoseconds(wait_ns));
    }
    return true;
  }

 private:
  std::array<T, CAP> buf_{};
  alignas(64) std::atomic<size_t> w_{0};
  alignas(64) std::atomic<size_t> r_{0};
};

// ---------------- thread affinity + naming ----------------

inline bool set_thread_affinity(int cpu) {
  cpu_set_t set;
  CPU_ZERO(&set);
  CPU_SET(cpu, &set);
  return pthread_setaffinity_np(pthread_self(), sizeof(set), &set) == 0;
}

inline void set_thread_name(const char* name) {
  pthread_setname_np(pthread_self(), name);  // <=15 chars
}

// ---------------- async write pool ----------------
// The reference posts baseband/spectrum/time-series product writes onto
// boost::asio::thread_pools (write_signal_pipe.hpp:55-57,159,210,249) so a
// detection dump never stalls the streaming pipeline; this is the plain-C++
// equivalent.  Destructor drains the queue before joining (writes are never
// lost at shutdown).
class WritePool {
 public:
  explicit WritePool(int n_threads = 2) {
    for (int i = 0; i < n_threads; ++i)
      threads_.emplace_back([this] {
        set_thread_name("srtb-writer");
        run();
      });
  }
  WritePool(const WritePool&) = delete;
  WritePool& operator=(const WritePool&) = delete;

  ~WritePool() {
    {
      std::unique_lock<std::mutex> lk(m_);
      done_ = true;
    }
    cv_.notify_all();
    for (auto& t : threads_) t.join();
  }

  void post(std::function<void()> f) {
    {
      std::unique_lock<std::mutex> lk(m_);
      q_.push_back(std::move(f));
    }
    cv_.notify_one();
  }

  // Block until every posted task has finished (for tests / clean exits
  // that must observe the files).
  void drain() {
    std::unique_lock<std::mutex> lk(m_);
    idle_cv_.wait(lk, [this] { return q_.empty() && active_ == 0; });
  }

 private:
  void run() {
    for (;;) {
      std::function<void()> f;
      {
        std::unique_lock<std::mutex> lk(m_);
        cv_.wait(lk, [this] { return done_ || !q_.empty(); });
        if (q_.empty()) {
          if (done_) return;
          continue;
        }
        f = std::move(q_.front());
        q_.pop_front();
        ++active_;
      }
      try {
        f();
      } catch (const std::exception& e) {
        SRTB_APP_LOGE("write pool: " << e.what());
      }
      {
        std::unique_lock<std::mutex> lk(m_);
        --active_;
        if (q_.empty() && active_ == 0) idle_cv_.notify_all();
      }
    }
  }

  std::mutex m_;
  std::condition_variable cv_, idle_cv_;
  std::deque<std::function<void()>> q_;
  int active_ = 0;
  bool done_ = false;
  std::vector<std::thread> threads_;
};

}  // namespace srtb_app
