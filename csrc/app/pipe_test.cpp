// Unit test of the generic pipe framework (csrc/app/pipe.h) — a CPU-only
// executable run by tests/test_native_app.py.  Builds a 3-stage pipeline
// with bounded queues + a composite pipe + a loose (drop-when-full) branch
// and checks ordering, backpressure and clean shutdown.
#include <cstdio>
#include <numeric>
#include <vector>

#include "pipe.h"
#include "runtime.h"

using namespace srtb_app;

namespace {

struct Work {
  int id = 0;
  long value = 0;
};

int failures = 0;
#define CHECK(cond)                                                       \
  do {                                                                    \
    if (!(cond)) {                                                        \
      std::printf("CHECK failed at %s:%d: %s\n", __FILE__, __LINE__,      \
                  #cond);                                                 \
      ++failures;                                                         \
    }                                                                     \
  } while (0)

void test_three_stage_pipeline() {
  constexpr int kN = 1000;
  SpscQueue<Work, 2> q1;  // depth-2 queues → real backpressure
  SpscQueue<Work, 2> q2;
  StopFlag stop;

  std::vector<long> got;
  // stage 1: generator (dummy in; pushes kN works then stops its pipe)
  auto gen_in = [n = 0](const StopFlag&) mutable -> std::optional<int> {
    if (n >= kN) return std::nullopt;
    return n++;
  };
  auto gen_fn = [](const StopFlag&, int i) -> std::optional<Work> {
    return Work{i, (long)i};
  };
  // stage 2: transform x -> 3x + 1
  auto tr_fn = [](const StopFlag&, Work w) -> std::optional<Work> {
    w.value = 3 * w.value + 1;
    return w;
  };
  // stage 3: collector
  auto sink_fn = [&got](const StopFlag&, Work w) -> std::optional<int> {
    got.push_back(w.value);
    return 0;
  };
  auto null_out = [](const StopFlag&, int) {};

  auto t1 = start_pipe("gen", gen_fn, gen_in,
                       QueueOut<SpscQueue<Work, 2>>{&q1}, stop);
  auto t2 = start_pipe("transform", tr_fn, QueueIn<SpscQueue<Work, 2>>{&q1},
                       QueueOut<SpscQueue<Work, 2>>{&q2}, stop);
  auto t3 = start_pipe("sink", sink_fn, QueueIn<SpscQueue<Work, 2>>{&q2},
                       null_out, stop);
  t1.join();
  // generator stopped; let the tail drain then stop the rest
  while (got.size() < kN) std::this_thread::yield();
  stop.request_stop();
  t2.join();
  t3.join();

  CHECK((int)got.size() == kN);
  for (int i = 0; i < kN; ++i) CHECK(got[i] == 3L * i + 1);
}

void test_composite_pipe() {
  StopFlag stop;
  auto add = [](const StopFlag&, long v) -> std::optional<long> {
    return v + 10;
  };
  auto mul = [](const StopFlag&, long v) -> std::optional<long> {
    return v * 2;
  };
  auto drop_odd = [](const StopFlag&, long v) -> std::optional<long> {
    if (v % 2) return std::nullopt;
    return v;
  };
  auto c = compose(add, mul, drop_odd);
  auto r = c(stop, 5L);
  CHECK(r && *r == 30);  // (5+10)*2, even → passes
  auto c2 = compose(add, drop_odd);
  CHECK(!c2(stop, 5L));  // 15 is odd → composite short-circuits
}

void test_loose_out_drops_under_load() {
  SpscQueue<int, 2> q;
  StopFlag stop;
  LooseQueueOut<SpscQueue<int, 2>> out{&q};
  for (int i = 0; i < 100; ++i) out(stop, int(i));
  // queue keeps only the first 2 (nobody popped); the rest were dropped
  int a = -1, b = -1, c = -1;
  CHECK(q.try_pop(a) && a == 0);
  CHECK(q.try_pop(b) && b == 1);
  CHECK(!q.try_pop(c));
}

void test_stop_unblocks_waiters() {
  SpscQueue<int, 2> q;
  StopFlag stop;
  auto in = QueueIn<SpscQueue<int, 2>>{&q};
  std::thread t([&] {
    auto v = in(stop);  // blocks: queue stays empty
    CHECK(!v);          // stop → nullopt, not a value
  });
  std::this_thread::sleep_for(std::chrono::milliseconds(50));
  stop.request_stop();
  t.join();
}

}  // namespace

int main() {
  test_three_stage_pipeline();
  test_composite_pipe();
  test_loose_out_drops_under_load();
  test_stop_unblocks_waiters();
  if (failures) {
    std::printf("PIPE TEST FAILED (%d)\n", failures);
    return 1;
  }
  std::printf("PIPE TEST OK\n");
  return 0;
}
