// Generic pipe framework — the reference's pipeline/framework re-designed
// for the MI355X app layer (reference framework/pipe.hpp:100-175,
// pipe_io.hpp:27-152, composite_pipe.hpp:28-68).
//
// A pipe functor has the signature
//     std::optional<Out> operator()(const StopFlag&, In)
// and runs on its own named thread in a pop → transform → push loop over
// bounded SPSC queues; returning std::nullopt ends the pipe.  Differences
// from the reference are deliberate: std::thread + a shared StopFlag
// instead of C++20 jthread/stop_token (HIP toolchain is C++17 here), and
// queues are explicit constructor arguments instead of globals.
#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <optional>
#include <string>
#include <thread>
#include <utility>

#include "runtime.h"

namespace srtb_app {

// cooperative stop signal shared by every pipe of one pipeline
class StopFlag {
 public:
  StopFlag() : flag_(std::make_shared<std::atomic<bool>>(false)) {}
  void request_stop() const { flag_->store(true, std::memory_order_relaxed); }
  bool stop_requested() const {
    return flag_->load(std::memory_order_relaxed);
  }
  bool operator()() const { return stop_requested(); }

 private:
  std::shared_ptr<std::atomic<bool>> flag_;
};

// ---- IO functors over bounded SPSC queues (reference pipe_io.hpp) ----

template <typename Q>
struct QueueIn {
  Q* q;
  using value_type = typename Q::value_type;
  std::optional<value_type> operator()(const StopFlag& stop) {
    value_type v;
    if (!q->pop(v, stop)) return std::nullopt;
    return v;
  }
};

template <typename Q>
struct QueueOut {
  Q* q;
  template <typename T>
  void operator()(const StopFlag& stop, T&& v) {
    q->push(std::forward<T>(v), stop);
  }
};

// drop-when-full push (the reference's loose_queue_out_functor — GUI branch
// drops frames under load, pipe_io.hpp:79-94)
template <typename Q>
struct LooseQueueOut {
  Q* q;
  template <typename T>
  void operator()(const StopFlag&, T&& v) {
    (void)q->try_push(std::forward<T>(v));
  }
};

// fan one work out to several queues (pipe_io.hpp:97-112); the work type
// must be copyable
template <typename Q1, typename Q2>
struct FanOut2 {
  Q1* q1;
  Q2* q2;
  template <typename T>
  void operator()(const StopFlag& stop, const T& v) {
    q1->push(v, stop);
    q2->push(v, stop);
  }
};

// ---- the pipe loop (reference pipe.hpp:108-141) ----

template <typename Fn, typename InF, typename OutF>
void run_pipe(Fn fn, InF in, OutF out, StopFlag stop) {
  while (!stop.stop_requested()) {
    auto opt_in = in(stop);
    if (!opt_in || stop.stop_requested()) break;
    auto opt_out = fn(stop, std::move(*opt_in));
    if (!opt_out || stop.stop_requested()) break;
    out(stop, std::move(*opt_out));
  }
}

// spawn a named pipe thread (reference start_pipe, pipe.hpp:149-175)
template <typename Fn, typename InF, typename OutF>
std::thread start_pipe(const char* name, Fn fn, InF in, OutF out,
                       StopFlag stop) {
  std::thread t([name, fn = std::move(fn), in = std::move(in),
                 out = std::move(out), stop]() mutable {
    set_thread_name(name);
    run_pipe(std::move(fn), std::move(in), std::move(out), stop);
  });
  return t;
}

// ---- composite pipe: chain functors inside ONE thread ----
// (reference composite_pipe.hpp:28-68; used by srtb-baseband-receiver)

template <typename F1, typename F2>
struct CompositePipe {
  F1 f1;
  F2 f2;
  template <typename In>
  auto operator()(const StopFlag& stop, In&& in)
      -> decltype(f2(stop, std::move(*f1(stop, std::forward<In>(in))))) {
    auto mid = f1(stop, std::forward<In>(in));
    if (!mid) return std::nullopt;
    return f2(stop, std::move(*mid));
  }
};

template <typename F1, typename F2>
CompositePipe<F1, F2> compose(F1 f1, F2 f2) {
  return {std::move(f1), std::move(f2)};
}

template <typename F1, typename F2, typename F3, typename... Fs>
auto compose(F1 f1, F2 f2, F3 f3, Fs... fs) {
  return compose(CompositePipe<F1, F2>{std::move(f1), std::move(f2)},
                 std::move(f3), std::move(fs)...);
}

}  // namespace srtb_app
