// srtb-backend — the native MI355X real-time radio-telescope backend
// executable (reference src/main.cpp:61-333 equivalent).
//
// Pipeline: input thread(s) (file replay with dedispersion-overlap seek-back,
// or one recvmmsg UDP ingest thread per configured endpoint, pinned to the
// configured cores) → bounded queue of pinned block buffers → GPU engines
// (double-buffered HIP streams; formats with 2 data streams fan one packed
// block out into per-polarization engines on a device-side unpack stream,
// reference unpack_pipe.hpp:146-390) → detection gate + cross-polarization
// coincidence (reference write_signal_pipe.hpp:81-140) → async product
// dumps (D2H staged on the slot stream, files written on a pool).
//
// Extra (non-reference) flags: --max-blocks N, --dry-run (parse config and
// print the resolved values — used by CPU tests), --print-config.

#include <hip/hip_runtime.h>

#include <atomic>
#include <cinttypes>
#include <cmath>
#include <condition_variable>
#include <cstring>
#include <deque>
#include <fstream>
#include <memory>
#include <mutex>
#include <numeric>
#include <thread>
#include <vector>

#include "../engine/engine.h"
#include "comm.h"
#include "config.h"
#include "runtime.h"
#include "udp_receiver.h"
#include "writers.h"

using namespace srtb_app;
using srtb_hip::EngineConfig;
using srtb_hip::PipelineEngine;

namespace {

// reference coherent_dedispersion.hpp:87-128 (Python twin: ref.nsamps_reserved)
size_t nsamps_reserved(const Config& c) {
  if (!c.baseband_reserve_sample) return 0;
  const double f = c.baseband_freq_low + c.baseband_bandwidth;
  const double fc = c.baseband_freq_low;
  const double delay =
      -4.148808e3 * c.dm * (1.0 / (f * f) - 1.0 / (fc * fc));
  const long long minimal =
      2 * (long long)std::llround(delay * c.baseband_sample_rate);
  const long long per_bin = 2 * (long long)c.spectrum_channel_count;
  const long long n = (long long)c.baseband_input_count;
  const long long refft_total = (n - minimal) / per_bin * per_bin;
  if (refft_total <= 0) {
    SRTB_APP_LOGW("nsamps_reserved > baseband_input_count; overlap disabled");
    return 0;
  }
  return (size_t)(n - refft_total);
}

// reference spectrum/rfi_mitigation.hpp:63-157 (Python twin: ref.rfi_*)
void parse_zap_ranges(const Config& c, EngineConfig& e) {
  e.n_zap_ranges = 0;
  if (c.mitigate_rfi_freq_list.empty()) return;
  const size_t nc = c.baseband_input_count / 2;
  std::stringstream ss(c.mitigate_rfi_freq_list);
  std::string part;
  while (std::getline(ss, part, ',') && e.n_zap_ranges < 16) {
    part = trim(part);
    const size_t dash = part.find('-');
    if (dash == std::string::npos) continue;
    double lo = std::stod(part.substr(0, dash));
    double hi = std::stod(part.substr(dash + 1));
    const bool bw_neg = c.baseband_bandwidth < 0;
    if (((hi - lo) < 0) != bw_neg) std::swap(lo, hi);
    const long long i_lo = std::llround((lo - c.baseband_freq_low) /
                                        c.baseband_bandwidth * (nc - 1));
    const long long i_hi = std::llround((hi - c.baseband_freq_low) /
                                        c.baseband_bandwidth * (nc - 1));
    if (0 <= i_lo && i_lo <= i_hi && (size_t)i_hi < nc) {
      e.zap_ranges[e.n_zap_ranges].lo = (unsigned long long)i_lo;
      e.zap_ranges[e.n_zap_ranges].hi = (unsigned long long)i_hi;
      ++e.n_zap_ranges;
    } else {
      SRTB_APP_LOGW("RFI range " << lo << "-" << hi << " MHz out of band");
    }
  }
}

struct BlockMsg {
  int buf_index = -1;
  uint64_t counter = 0;
  bool eof = false;
};

// multi-producer free list of pinned block buffers (writer threads return
// buffers too, so the SPSC queue is not enough here)
class FreeList {
 public:
  void put(int i) {
    {
      std::lock_guard<std::mutex> l(m_);
      v_.push_back(i);
    }
    cv_.notify_one();
  }
  template <typename Stop>
  bool get(int& out, Stop stop) {
    std::unique_lock<std::mutex> l(m_);
    while (v_.empty()) {
      if (stop()) return false;
      cv_.wait_for(l, std::chrono::milliseconds(2));
    }
    out = v_.back();
    v_.pop_back();
    return true;
  }

 private:
  std::mutex m_;
  std::condition_variable cv_;
  std::vector<int> v_;
};

// Per-(engine,slot) pinned staging for async detection dumps: waterfall,
// time series and cumulative sum go D2H on the SLOT stream (ordered before
// any next block's kernels on that stream), an event marks completion, and
// the file writes happen on the WritePool — the submit thread never blocks
// on a 4 GB hipMemcpy (reference posts product writes to asio pools).
struct DumpStaging {
  std::complex<float>* wf = nullptr;
  float* ts = nullptr;
  float* cumsum = nullptr;
  hipEvent_t ev = nullptr;
  std::atomic<bool> busy{false};
};

}  // namespace

int main(int argc, char** argv) {
  install_termination_handler();

  // strip runner-only flags before config parsing
  long long max_blocks = -1;
  bool dry_run = false;
  std::vector<char*> cfg_argv{argv[0]};
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--max-blocks") {
      max_blocks = std::stoll(argv[++i]);
    } else if (a.rfind("--max-blocks=", 0) == 0) {
      max_blocks = std::stoll(a.substr(13));
    } else if (a == "--dry-run" || a == "--print-config") {
      dry_run = true;
    } else {
      cfg_argv.push_back(argv[i]);
    }
  }

  Config cfg;
  try {
    cfg.parse_args((int)cfg_argv.size(), cfg_argv.data());
  } catch (const std::exception& e) {
    SRTB_APP_LOGE("config: " << e.what());
    return 2;
  }
  log_level().store(cfg.log_level);

  // polarization fan-out: 2-stream formats run one packed packet stream
  // through per-pol engines (reference backend_registry get_data_stream_count)
  const std::string fmt_name = cfg.baseband_format_type;
  const int n_streams =
      (fmt_name == "naocpsr_snap1" || fmt_name == "gznupsr_a1") ? 2 : 1;

  const size_t reserved_ref = nsamps_reserved(cfg);
  size_t reserved = reserved_ref;
  const bool udp_mode = cfg.input_file_path.empty();
  if (udp_mode && reserved > 0) {
    // UDP overlap reservation (the reference left this as a TODO,
    // udp_receiver_pipe.hpp:38): each block keeps the previous block's tail.
    // The reserved sample count must align to both the packet payload and
    // the 2*S valid-window rule, so round it UP to the lcm.
    const auto fmt = PacketFormat::for_backend(fmt_name);
    const int bits = std::abs(cfg.baseband_input_bits);
    const size_t samples_per_payload =
        fmt.payload_size * 8 / (size_t)bits / (size_t)n_streams;
    const size_t step =
        std::lcm(samples_per_payload, 2 * cfg.spectrum_channel_count);
    const size_t adj = (reserved + step - 1) / step * step;
    if (adj != reserved)
      SRTB_APP_LOGI("UDP overlap: nsamps_reserved rounded " << reserved
                    << " -> " << adj << " (payload/2S alignment)");
    reserved = adj;
  }

  if (dry_run) {
    std::printf("%s", cfg.dump().c_str());
    std::printf("data_stream_count = %d\n", n_streams);
    if (reserved != reserved_ref)
      std::printf("nsamps_reserved_udp = %zu\n", reserved);
    std::printf("nsamps_reserved = %zu\n", reserved_ref);
    return 0;
  }

  EngineConfig ec;
  ec.baseband_input_count = cfg.baseband_input_count;
  // fan-out formats carry int8 per-pol sample streams
  ec.baseband_input_bits = (n_streams > 1) ? -8 : cfg.baseband_input_bits;
  ec.spectrum_channel_count = cfg.spectrum_channel_count;
  ec.freq_low = cfg.baseband_freq_low;
  ec.bandwidth = cfg.baseband_bandwidth;
  ec.sample_rate = cfg.baseband_sample_rate;
  ec.dm = cfg.dm;
  ec.rfi_threshold = (float)cfg.mitigate_rfi_average_method_threshold;
  ec.sk_threshold = (float)cfg.mitigate_rfi_spectral_kurtosis_threshold;
  ec.snr_threshold = (float)cfg.signal_detect_signal_noise_threshold;
  ec.max_boxcar_length = cfg.signal_detect_max_boxcar_length;
  ec.nsamps_reserved = reserved;
  parse_zap_ranges(cfg, ec);

  // multi-GPU: one process per GPU (torchrun or shell-loop launched); the
  // communicator binds this rank's device BEFORE any engine allocation
  Comm comm;
  if (comm.world() > 1) {
    SRTB_APP_LOGI("multi-GPU: rank " << comm.rank() << "/" << comm.world()
                                     << " on device " << comm.local_rank());
    // per-rank product prefix so counters from different ranks never collide
    cfg.baseband_output_file_prefix += "r" + std::to_string(comm.rank()) + "_";
  }

  SRTB_APP_LOGI("srtb-backend: N=" << cfg.baseband_input_count
                                   << " bits=" << cfg.baseband_input_bits
                                   << " S=" << cfg.spectrum_channel_count
                                   << " dm=" << cfg.dm
                                   << " reserved=" << reserved
                                   << " streams=" << n_streams);

  std::vector<std::unique_ptr<PipelineEngine>> engines;
  for (int i = 0; i < n_streams; ++i)
    engines.emplace_back(std::make_unique<PipelineEngine>(ec, 2));
  PipelineEngine& engine = *engines[0];
  const size_t n_per_pol = cfg.baseband_input_count;
  const size_t raw_bytes = (n_streams > 1)
                               ? n_per_pol * (size_t)n_streams
                               : engine.raw_bytes();

  // device-side fan-out resources: the packed block uploads + unpacks on a
  // dedicated stream; engines wait on the recorded event (no host sync)
  hipStream_t fan_stream = nullptr;
  struct FanRing {
    uint8_t* d_raw = nullptr;
    float* pol0 = nullptr;
    float* pol1 = nullptr;
    hipEvent_t ev = nullptr;
  };
  std::array<FanRing, 2> fan{};
  int fan_next = 0;
  if (n_streams > 1) {
    srtb_hip::check_hip(
        hipStreamCreateWithFlags(&fan_stream, hipStreamNonBlocking),
        "fan stream");
    for (auto& r : fan) {
      srtb_hip::check_hip(hipMalloc(&r.d_raw, raw_bytes), "fan raw");
      srtb_hip::check_hip(hipMalloc(&r.pol0, n_per_pol * sizeof(float)),
                          "fan pol0");
      srtb_hip::check_hip(hipMalloc(&r.pol1, n_per_pol * sizeof(float)),
                          "fan pol1");
      srtb_hip::check_hip(
          hipEventCreateWithFlags(&r.ev, hipEventDisableTiming), "fan ev");
    }
  }

  // ring of pinned block buffers feeding the engines; writer threads return
  // buffers after async dumps, so release is multi-producer
  const int n_endpoints =
      udp_mode ? std::max<int>(1, (int)cfg.udp_receiver_address.size()) : 1;
  const int kBufs = 4 + 2 * (n_endpoints - 1);
  std::vector<uint8_t*> bufs(kBufs);
  for (auto& b : bufs)
    srtb_hip::check_hip(hipHostMalloc(&b, raw_bytes), "pinned block");
  SpscQueue<BlockMsg, 64> q_in;
  FreeList q_free;
  for (int i = 0; i < kBufs; ++i) q_free.put(i);
  std::atomic<bool> stop{false};
  std::atomic<long long> blocks_in{0};
  auto stopped = [&] { return stop.load(std::memory_order_relaxed); };

  std::vector<std::thread> input_threads;
  // shared with the input threads, which outlive any branch-local scope
  // (capturing block-locals by reference here was a use-after-scope that
  // corrupted main's stack — found by ASan)
  std::atomic<int> udp_eof_count{0};
  std::vector<int> udp_my_eps;
  int udp_n_my_eps = 0;
  if (!udp_mode) {
    // ---- file replay with overlap seek-back (read_file_pipe.hpp:58-126) ----
    input_threads.emplace_back([&] {
      set_thread_name("srtb_input");
      std::ifstream f(cfg.input_file_path, std::ios::binary);
      if (!f) {
        SRTB_APP_LOGE("cannot open " << cfg.input_file_path);
        stop.store(true);
        return;
      }
      const int bits = std::abs(cfg.baseband_input_bits);
      size_t res_bytes = reserved * (size_t)n_streams * bits / 8;
      if (res_bytes >= raw_bytes) res_bytes = 0;
      uint64_t pos = cfg.input_file_offset_bytes +
                     (uint64_t)comm.rank() * (raw_bytes - res_bytes);
      long long count = 0;
      while (!stopped()) {
        if (max_blocks >= 0 && count >= max_blocks) break;
        int bi;
        if (!q_free.get(bi, stopped)) break;
        f.seekg((std::streamoff)pos);
        f.read(reinterpret_cast<char*>(bufs[bi]), (std::streamsize)raw_bytes);
        if ((size_t)f.gcount() < raw_bytes) {
          q_free.put(bi);
          break;  // EOF
        }
        const uint64_t sample_index = pos * 8 / bits;
        q_in.push(BlockMsg{bi, sample_index, false}, stopped);
        // multi-GPU file replay: rank r processes blocks r, r+W, 2W, ...
        pos += (uint64_t)(raw_bytes - res_bytes) * (uint64_t)comm.world();
        ++count;
      }
      q_in.push(BlockMsg{-1, 0, true}, stopped);
    });
  } else {
    // ---- UDP ingest: one recvmmsg receiver per endpoint (reference
    // main.cpp:230-272 spawns N receiver pipes), each pinned to its core ----
    const int bits = std::abs(cfg.baseband_input_bits);
    size_t res_bytes = reserved * (size_t)n_streams * bits / 8;
    if (res_bytes >= raw_bytes) res_bytes = 0;
    udp_my_eps = shard_indices(n_endpoints, comm.world(), comm.rank());
    udp_n_my_eps = (int)udp_my_eps.size();
    if (udp_n_my_eps == 0) {
      SRTB_APP_LOGW("rank " << comm.rank() << ": no UDP endpoints assigned");
      q_in.push(BlockMsg{-1, 0, true}, stopped);
    }
    for (const int ep : udp_my_eps) {
      input_threads.emplace_back([&, ep, res_bytes, bits] {
        set_thread_name(("srtb_udp" + std::to_string(ep)).c_str());
        if ((size_t)ep < cfg.udp_receiver_cpu_preferred.size())
          set_thread_affinity(cfg.udp_receiver_cpu_preferred[ep]);
        try {
          auto fmt = PacketFormat::for_backend(fmt_name);
          RecvmmsgProvider prov(cfg.udp_receiver_address.at(ep),
                                cfg.udp_receiver_port.at(ep),
                                fmt.packet_size());
          int bi;
          if (!q_free.get(bi, stopped)) return;
          // assembler fills the NEW region after the overlap head; the
          // previous block's tail is copied in front on completion
          BlockAssembler assembler(fmt, raw_bytes - res_bytes,
                                   bufs[bi] + res_bytes);
          std::vector<uint8_t> prev_tail(res_bytes, 0);
          long long count = 0;
          uint64_t next_log = 1;
          bool done = false;
          while (!stopped() && !done) {
            const int got = prov.receive();
            for (int i = 0; i < got && !done; ++i) {
              if (assembler.push(prov.packet(i), prov.packet_len(i))) {
                if (res_bytes) {
                  std::memcpy(bufs[bi], prev_tail.data(), res_bytes);
                  std::memcpy(prev_tail.data(),
                              bufs[bi] + raw_bytes - res_bytes, res_bytes);
                }
                q_in.push(BlockMsg{bi, assembler.block_begin_counter(), false},
                          stopped);
                ++count;
                const auto& st = assembler.stats();
                if ((uint64_t)count >= next_log) {
                  SRTB_APP_LOGI("udp[" << ep << "] block "
                                << assembler.block_begin_counter()
                                << ": received=" << st.received
                                << " lost=" << st.lost << " dup="
                                << st.duplicate << " loss_rate="
                                << st.loss_rate());
                  next_log *= 2;  // log blocks 1,2,4,8,... then every 256
                  if (next_log > 256) next_log = (uint64_t)count + 256;
                }
                if (max_blocks >= 0 && count >= max_blocks) {
                  done = true;  // per-endpoint block budget reached
                  break;
                }
                if (!q_free.get(bi, stopped)) return;
                assembler.set_block_buffer(bufs[bi] + res_bytes);
                assembler.begin_next();
              }
            }
          }
          blocks_in.fetch_add(count);
        } catch (const std::exception& e) {
          SRTB_APP_LOGE("udp[" << ep << "]: " << e.what());
          stop.store(true);
        }
        if (udp_eof_count.fetch_add(1) + 1 == udp_n_my_eps)
          q_in.push(BlockMsg{-1, 0, true}, stopped);
      });
    }
  }

  // ---- GPU pipeline + writer loop ----
  struct InFlight {
    std::array<int, 2> slots{-1, -1};
    int buf_index;
    uint64_t counter;
  };
  std::vector<InFlight> inflight;
  const size_t S = engine.n_channels(), Lw = engine.waterfall_len();
  const size_t ts_count = engine.ts_count();
  uint64_t blocks = 0, detections = 0;
  srtb_app::WritePool writers(2);
  const std::string out_prefix = cfg.baseband_output_file_prefix;

  // baseband_write_all: record every block minus the overlap tail into one
  // rolling file per rank (the reference replaces the write_signal tail
  // with write_file_pipe, write_file_pipe.hpp:41-94).  Appends go to the
  // pool as pwrite-at-offset so two writer threads can't reorder them.
  int write_all_fd = -1;
  size_t wa_valid = 0;
  uint64_t wa_index = 0;
  if (cfg.baseband_write_all) {
    size_t res_b = reserved * (size_t)n_streams *
                   (size_t)std::abs(cfg.baseband_input_bits) / 8;
    if (res_b >= raw_bytes) res_b = 0;
    wa_valid = raw_bytes - res_b;
    const std::string path = out_prefix + "all_r" +
                             std::to_string(comm.rank()) + ".bin";
    write_all_fd = ::open(path.c_str(), O_WRONLY | O_CREAT | O_TRUNC, 0644);
    if (write_all_fd < 0) {
      SRTB_APP_LOGE("cannot open " << path);
      return 3;
    }
    SRTB_APP_LOGI("baseband_write_all -> " << path << " (" << wa_valid
                                           << " valid bytes/block)");
  }

  // async dump staging, lazily pinned on first detection
  std::vector<std::array<DumpStaging, 2>> staging(n_streams);

  // cross-pol / cross-block coincidence state (write_signal_pipe.hpp:81-140):
  // counters of recent positive blocks; one held-back negative block buffer
  std::deque<uint64_t> recent_positive;
  uint64_t block_delta = 0, last_counter = 0;
  bool have_last = false;
  int held_buf = -1;
  uint64_t held_counter = 0;

  auto coincident = [&](uint64_t c) {
    const double window = 0.45 * (double)block_delta;
    for (uint64_t t : recent_positive)
      if (std::abs((double)(int64_t)(c - t)) <= window) return true;
    return false;
  };

  auto submit_block = [&](int bi) -> std::array<int, 2> {
    std::array<int, 2> slots{-1, -1};
    if (n_streams == 1) {
      slots[0] = engine.submit(bufs[bi], raw_bytes);
      return slots;
    }
    FanRing& r = fan[fan_next];
    fan_next ^= 1;
    srtb_hip::check_hip(hipMemcpyAsync(r.d_raw, bufs[bi], raw_bytes,
                                       hipMemcpyHostToDevice, fan_stream),
                        "fan h2d");
    if (fmt_name == "gznupsr_a1")
      srtb_hip::check_hip(
          srtb_hip::unpack_gznupsr_a1(r.d_raw, r.pol0, r.pol1, nullptr,
                                      nullptr, 2, n_per_pol, nullptr,
                                      fan_stream),
          "fan gznupsr");
    else
      srtb_hip::check_hip(
          srtb_hip::unpack_naocpsr_snap1(
              reinterpret_cast<const int8_t*>(r.d_raw), r.pol0, r.pol1,
              n_per_pol, nullptr, fan_stream),
          "fan snap1");
    srtb_hip::check_hip(hipEventRecord(r.ev, fan_stream), "fan ev rec");
    slots[0] = engines[0]->submit_samples_device(r.pol0, n_per_pol, NAN, r.ev);
    slots[1] = engines[1]->submit_samples_device(r.pol1, n_per_pol, NAN, r.ev);
    return slots;
  };
  auto dump_stream = [&](int si, int slot, const srtb_hip::BlockResult& res,
                         uint64_t counter) {
    PipelineEngine& eng = *engines[si];
    DumpStaging& st = staging[si][slot];
    while (st.busy.load(std::memory_order_acquire))
      std::this_thread::yield();  // previous dump of this slot still writing
    if (!st.wf) {
      srtb_hip::check_hip(
          hipHostMalloc(&st.wf, S * Lw * sizeof(std::complex<float>)),
          "stage wf");
      srtb_hip::check_hip(hipHostMalloc(&st.ts, ts_count * sizeof(float)),
                          "stage ts");
      srtb_hip::check_hip(hipHostMalloc(&st.cumsum, ts_count * sizeof(float)),
                          "stage cumsum");
      srtb_hip::check_hip(hipEventCreateWithFlags(&st.ev,
                                                  hipEventDisableTiming),
                          "stage ev");
    }
    hipStream_t strm = eng.stream(slot);
    srtb_hip::check_hip(hipMemcpyAsync(st.wf, eng.waterfall_ptr(slot),
                                       S * Lw * sizeof(float2),
                                       hipMemcpyDeviceToHost, strm),
                        "wf d2h");
    srtb_hip::check_hip(hipMemcpyAsync(st.ts, eng.time_series_ptr(slot),
                                       ts_count * sizeof(float),
                                       hipMemcpyDeviceToHost, strm),
                        "ts d2h");
    srtb_hip::check_hip(hipMemcpyAsync(st.cumsum, eng.cumsum_ptr(slot),
                                       ts_count * sizeof(float),
                                       hipMemcpyDeviceToHost, strm),
                        "cumsum d2h");
    srtb_hip::check_hip(hipEventRecord(st.ev, strm), "stage ev rec");
    st.busy.store(true, std::memory_order_release);
    auto counts = res.counts;
    DumpStaging* stp = &st;
    const size_t S_ = S, Lw_ = Lw, tsc = ts_count;
    const std::string prefix = out_prefix;
    writers.post([stp, counts, counter, S_, Lw_, tsc, prefix] {
      srtb_hip::check_hip(hipEventSynchronize(stp->ev), "stage ev sync");
      write_spectrum_npy(prefix, counter, stp->wf, S_, Lw_);
      for (const auto& [len, cnt] : counts) {
        if (cnt == 0) continue;
        if (len == 1) {
          write_time_series_tim(prefix, counter, 1, stp->ts, tsc);
        } else {
          const size_t n_out = tsc - len;
          std::vector<float> box(n_out);
          for (size_t i = 0; i < n_out; ++i)
            box[i] = stp->cumsum[i + len] - stp->cumsum[i];
          write_time_series_tim(prefix, counter, len, box.data(), n_out);
        }
      }
      stp->busy.store(false, std::memory_order_release);
    });
  };

  // write a block's raw baseband async and release its pinned buffer after
  auto write_raw_and_release = [&](int bi, uint64_t counter) {
    uint8_t* p = bufs[bi];
    const size_t nb = raw_bytes;
    const std::string prefix = out_prefix;
    writers.post([&q_free, p, nb, bi, counter, prefix] {
      write_baseband_bin(prefix, counter, p, nb);
      q_free.put(bi);
    });
  };

  auto drain_one = [&] {
    const InFlight w = inflight.front();
    inflight.erase(inflight.begin());
    std::vector<srtb_hip::BlockResult> res(n_streams);
    for (int si = 0; si < n_streams; ++si)
      res[si] = engines[si]->wait(w.slots[si]);
    ++blocks;
    if (write_all_fd >= 0) {
      // record-everything tail: append the valid region and release the
      // buffer from the pool; detections are still counted/logged below
      // by the normal gate, but product dumps are replaced by the file
      const uint64_t off = wa_index++ * (uint64_t)wa_valid;
      uint8_t* p = bufs[w.buf_index];
      const int bi = w.buf_index;
      const size_t nb = wa_valid;
      const int fd = write_all_fd;
      writers.post([&q_free, p, off, nb, bi, fd] {
        size_t done = 0;
        while (done < nb) {
          const ssize_t k =
              ::pwrite(fd, p + done, nb - done, (off_t)(off + done));
          if (k <= 0) break;
          done += (size_t)k;
        }
        q_free.put(bi);
      });
      bool any = false;
      for (int si = 0; si < n_streams; ++si)
        for (auto& [len, cnt] : res[si].counts) any |= cnt > 0;
      if (any) ++detections;
      return;
    }
    if (have_last && w.counter > last_counter)
      block_delta = w.counter - last_counter;
    last_counter = w.counter;
    have_last = true;

    bool any_positive = false;
    std::vector<bool> positive(n_streams, false);
    for (int si = 0; si < n_streams; ++si) {
      const bool gate =
          res[si].zero_count <
          cfg.signal_detect_channel_threshold * (double)S;
      uint64_t pos = 0;
      for (auto& [len, cnt] : res[si].counts) pos += cnt;
      positive[si] = gate && pos > 0;
      any_positive |= positive[si];
    }

    if (any_positive) {
      ++detections;
      for (int si = 0; si < n_streams; ++si) {
        if (!positive[si]) continue;
        uint64_t pos = 0;
        for (auto& [len, cnt] : res[si].counts) pos += cnt;
        SRTB_APP_LOGI("detection in block " << w.counter << " stream " << si
                                            << " (" << pos
                                            << " samples over threshold)");
        dump_stream(si, w.slots[si], res[si], w.counter);
      }
      recent_positive.push_back(w.counter);
      while (recent_positive.size() > 5) recent_positive.pop_front();
      // a held-back negative block coincident with this positive gets its
      // baseband dumped too (reference pending-negative re-check)
      if (held_buf >= 0 && coincident(held_counter)) {
        write_raw_and_release(held_buf, held_counter);
        held_buf = -1;
      }
      write_raw_and_release(w.buf_index, w.counter);
    } else if (coincident(w.counter)) {
      write_raw_and_release(w.buf_index, w.counter);
    } else {
      // hold this negative block's buffer back one round so a positive in
      // the NEXT block (±0.45 window) can still dump it
      if (held_buf >= 0) q_free.put(held_buf);
      held_buf = w.buf_index;
      held_counter = w.counter;
    }
  };

  while (true) {
    BlockMsg msg;
    if (!q_in.pop(msg, stopped)) break;
    if (msg.eof) break;
    inflight.push_back({submit_block(msg.buf_index), msg.buf_index,
                        msg.counter});
    if (inflight.size() >= 2) drain_one();
  }
  while (!inflight.empty()) drain_one();
  if (held_buf >= 0) q_free.put(held_buf);
  if (write_all_fd >= 0) {
    writers.drain();  // all appends on disk before fdatasync+close
    ::fdatasync(write_all_fd);
    ::close(write_all_fd);
  }
  stop.store(true);
  for (auto& t : input_threads)
    if (t.joinable()) t.join();
  for (auto& e : engines) e->synchronize();
  writers.drain();  // all product files on disk before the summary line
  for (auto& st_arr : staging)
    for (auto& st : st_arr) {
      if (st.wf) (void)hipHostFree(st.wf);
      if (st.ts) (void)hipHostFree(st.ts);
      if (st.cumsum) (void)hipHostFree(st.cumsum);
      if (st.ev) (void)hipEventDestroy(st.ev);
    }
  for (auto& r : fan) {
    if (r.d_raw) (void)hipFree(r.d_raw);
    if (r.pol0) (void)hipFree(r.pol0);
    if (r.pol1) (void)hipFree(r.pol1);
    if (r.ev) (void)hipEventDestroy(r.ev);
  }
  if (fan_stream) (void)hipStreamDestroy(fan_stream);
  for (auto b : bufs) (void)hipHostFree(b);

  SRTB_APP_LOGI("done: " << blocks << " blocks, " << detections
                         << " with detections");
  if (comm.active()) {
    // cluster-wide detection statistics over RCCL (xGMI)
    std::vector<uint64_t> stats{blocks, detections};
    comm.allreduce_sum(stats);
    if (comm.rank() == 0)
      std::printf("[srtb-backend] world=%d blocks=%" PRIu64
                  " detections=%" PRIu64 " (all ranks)\n",
                  comm.world(), stats[0], stats[1]);
  }
  std::printf("[srtb-backend] blocks=%" PRIu64 " detections=%" PRIu64 "\n",
              blocks, detections);
  return 0;
}
