// srtb-backend — the native MI355X real-time radio-telescope backend
// executable (reference src/main.cpp:61-333 equivalent).
//
// Pipeline: input thread (file replay with dedispersion-overlap seek-back,
// or recvmmsg UDP ingest pinned to the configured core) → bounded SPSC queue
// of pinned block buffers → GPU engine (double-buffered HIP streams) →
// detection gate → product writers (.bin/.npy/.tim).
//
// Extra (non-reference) flags: --max-blocks N, --dry-run (parse config and
// print the resolved values — used by CPU tests), --print-config.

#include <hip/hip_runtime.h>

#include <atomic>
#include <cinttypes>
#include <cmath>
#include <cstring>
#include <fstream>
#include <thread>
#include <vector>

#include "../engine/engine.h"
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

  const size_t reserved = nsamps_reserved(cfg);
  if (dry_run) {
    std::printf("%s", cfg.dump().c_str());
    std::printf("nsamps_reserved = %zu\n", reserved);
    return 0;
  }

  EngineConfig ec;
  ec.baseband_input_count = cfg.baseband_input_count;
  ec.baseband_input_bits = cfg.baseband_input_bits;
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

  SRTB_APP_LOGI("srtb-backend: N=" << cfg.baseband_input_count
                                   << " bits=" << cfg.baseband_input_bits
                                   << " S=" << cfg.spectrum_channel_count
                                   << " dm=" << cfg.dm
                                   << " reserved=" << reserved);

  PipelineEngine engine(ec, 2);
  const size_t raw_bytes = engine.raw_bytes();

  // ring of pinned block buffers feeding the engine
  constexpr int kBufs = 4;
  std::vector<uint8_t*> bufs(kBufs);
  for (auto& b : bufs)
    srtb_hip::check_hip(hipHostMalloc(&b, raw_bytes), "pinned block");
  SpscQueue<BlockMsg, kBufs> q_in;
  SpscQueue<int, kBufs> q_free;  // returned buffer indices
  for (int i = 0; i < kBufs; ++i) q_free.try_push(i);
  std::atomic<bool> stop{false};
  auto stopped = [&] { return stop.load(std::memory_order_relaxed); };

  std::thread input_thread;
  if (!cfg.input_file_path.empty()) {
    // ---- file replay with overlap seek-back (read_file_pipe.hpp:58-126) ----
    input_thread = std::thread([&] {
      set_thread_name("srtb_input");
      std::ifstream f(cfg.input_file_path, std::ios::binary);
      if (!f) {
        SRTB_APP_LOGE("cannot open " << cfg.input_file_path);
        stop.store(true);
        return;
      }
      const int bits = std::abs(cfg.baseband_input_bits);
      size_t res_bytes = reserved * bits / 8;
      if (res_bytes >= raw_bytes) res_bytes = 0;
      uint64_t pos = cfg.input_file_offset_bytes;
      long long count = 0;
      while (!stopped()) {
        if (max_blocks >= 0 && count >= max_blocks) break;
        int bi;
        if (!q_free.pop(bi, stopped)) break;
        f.seekg((std::streamoff)pos);
        f.read(reinterpret_cast<char*>(bufs[bi]), (std::streamsize)raw_bytes);
        if ((size_t)f.gcount() < raw_bytes) break;  // EOF
        const uint64_t sample_index = pos * 8 / bits;
        q_in.push(BlockMsg{bi, sample_index, false}, stopped);
        pos += raw_bytes - res_bytes;
        ++count;
      }
      q_in.push(BlockMsg{-1, 0, true}, stopped);
    });
  } else {
    // ---- UDP ingest (recvmmsg, pinned core) ----
    input_thread = std::thread([&] {
      set_thread_name("srtb_udp");
      if (!cfg.udp_receiver_cpu_preferred.empty())
        set_thread_affinity(cfg.udp_receiver_cpu_preferred[0]);
      try {
        auto fmt = PacketFormat::for_backend(cfg.baseband_format_type);
        RecvmmsgProvider prov(cfg.udp_receiver_address.at(0),
                              cfg.udp_receiver_port.at(0),
                              fmt.packet_size());
        int bi;
        if (!q_free.pop(bi, stopped)) return;
        BlockAssembler assembler(fmt, raw_bytes, bufs[bi]);
        long long count = 0;
        while (!stopped()) {
          const int got = prov.receive();
          for (int i = 0; i < got; ++i) {
            if (assembler.push(prov.packet(i), prov.packet_len(i))) {
              q_in.push(BlockMsg{bi, assembler.block_begin_counter(), false},
                        stopped);
              ++count;
              const auto& st = assembler.stats();
              SRTB_APP_LOGI("udp block " << assembler.block_begin_counter()
                            << ": received=" << st.received
                            << " lost=" << st.lost << " loss_rate="
                            << st.loss_rate());
              if (max_blocks >= 0 && count >= max_blocks) {
                q_in.push(BlockMsg{-1, 0, true}, stopped);
                return;
              }
              if (!q_free.pop(bi, stopped)) return;
              assembler.set_block_buffer(bufs[bi]);
              assembler.begin_next();
            }
          }
        }
      } catch (const std::exception& e) {
        SRTB_APP_LOGE("udp: " << e.what());
        stop.store(true);
      }
      q_in.push(BlockMsg{-1, 0, true}, stopped);
    });
  }

  // ---- GPU pipeline + writer loop ----
  struct InFlight { int slot; int buf_index; uint64_t counter; };
  std::vector<InFlight> inflight;
  const size_t S = engine.n_channels(), Lw = engine.waterfall_len();
  uint64_t blocks = 0, detections = 0;
  // product writes go to a small pool (reference posts them to asio
  // thread_pools) so a detection dump never stalls block submission;
  // device buffers are staged to host HERE (they recycle on next submit),
  // file IO + fdatasync happen on the pool
  srtb_app::WritePool writers(2);
  const std::string out_prefix = cfg.baseband_output_file_prefix;

  auto drain_one = [&] {
    const InFlight w = inflight.front();
    inflight.erase(inflight.begin());
    auto res = engine.wait(w.slot);
    ++blocks;
    const bool gate =
        res.zero_count <
        cfg.signal_detect_channel_threshold * (double)S;
    uint64_t positive = 0;
    for (auto& [len, cnt] : res.counts) positive += cnt;
    if (gate && positive > 0) {
      ++detections;
      SRTB_APP_LOGI("detection in block " << w.counter << " ("
                                          << positive << " samples over "
                                          << "threshold)");
      // raw baseband lives in a pinned buffer that recycles after the
      // q_free push below — copy it out for the async write
      {
        std::vector<uint8_t> bb(bufs[w.buf_index],
                                bufs[w.buf_index] + raw_bytes);
        writers.post([out_prefix, counter = w.counter,
                      bb = std::move(bb)] {
          write_baseband_bin(out_prefix, counter, bb.data(), bb.size());
        });
      }
      std::vector<std::complex<float>> h_wf(S * Lw);
      srtb_hip::check_hip(
          hipMemcpy(h_wf.data(), engine.waterfall_ptr(w.slot),
                    S * Lw * sizeof(float2), hipMemcpyDeviceToHost),
          "wf d2h");
      writers.post([out_prefix, counter = w.counter, S, Lw,
                    wf = std::move(h_wf)] {
        write_spectrum_npy(out_prefix, counter, wf.data(), S, Lw);
      });
      std::vector<float> h_ts(engine.ts_count());
      srtb_hip::check_hip(
          hipMemcpy(h_ts.data(), engine.time_series_ptr(w.slot),
                    engine.ts_count() * sizeof(float),
                    hipMemcpyDeviceToHost),
          "ts d2h");
      for (auto& [len, cnt] : res.counts) {
        if (cnt == 0) continue;
        if (len == 1) {
          writers.post([out_prefix, counter = w.counter, ts = h_ts] {
            write_time_series_tim(out_prefix, counter, 1, ts.data(),
                                  ts.size());
          });
        } else {
          float* box = engine.compute_boxcar(w.slot, len);
          std::vector<float> h_box(engine.ts_count() - len);
          srtb_hip::check_hip(hipMemcpy(h_box.data(), box,
                                        h_box.size() * sizeof(float),
                                        hipMemcpyDeviceToHost),
                              "box d2h");
          writers.post([out_prefix, counter = w.counter, len = len,
                        bx = std::move(h_box)] {
            write_time_series_tim(out_prefix, counter, len, bx.data(),
                                  bx.size());
          });
        }
      }
    }
    q_free.push(w.buf_index, stopped);
  };

  while (true) {
    BlockMsg msg;
    if (!q_in.pop(msg, stopped)) break;
    if (msg.eof) break;
    const int slot = engine.submit(bufs[msg.buf_index], raw_bytes);
    inflight.push_back({slot, msg.buf_index, msg.counter});
    if (inflight.size() >= 2) drain_one();
  }
  while (!inflight.empty()) drain_one();
  stop.store(true);
  if (input_thread.joinable()) input_thread.join();
  engine.synchronize();
  writers.drain();  // all product files on disk before the summary line
  for (auto b : bufs) (void)hipHostFree(b);

  SRTB_APP_LOGI("done: " << blocks << " blocks, " << detections
                         << " with detections");
  std::printf("[srtb-backend] blocks=%" PRIu64 " detections=%" PRIu64 "\n",
              blocks, detections);
  return 0;
}
