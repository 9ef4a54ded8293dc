#include "engine.h"

#include <roctracer/roctx.h>

#include <cmath>
#include <cstdlib>
#include <cstring>
#include <stdexcept>

namespace srtb_hip {

namespace {
constexpr double kD = 4.148808e3;  // dispersion constant (MHz^2 pc^-1 cm^3 s)

// rocTX ranges per pipeline stage (SRTB_ROCTX=1): the rocprofv3 marker
// domain shows unpack/fft/rfi/sk/detect spans per block — the reference's
// per-pipe thread names, in profiler form (SURVEY §5 tracing).
struct RoctxRange {
  bool on;
  explicit RoctxRange(const char* name)
      : on([] {
          static const bool enabled = [] {
            const char* e = std::getenv("SRTB_ROCTX");
            return e && std::atoi(e) != 0;
          }();
          return enabled;
        }()) {
    if (on) roctxRangePush(name);
  }
  ~RoctxRange() {
    if (on) roctxRangePop();
  }
};
}

PipelineEngine::PipelineEngine(const EngineConfig& cfg, int n_slots)
    : cfg_(cfg), n_slots_(n_slots) {
  n_ = cfg.baseband_input_count;
  nc_ = n_ / 2;
  s_ = cfg.spectrum_channel_count;
  if (s_ > nc_) s_ = nc_;  // reference watfft clamps (fft_pipe.hpp:300-306)
  l_ = nc_ / s_;
  const size_t reserved_bins = cfg.nsamps_reserved / s_;
  ts_count_ = (l_ > reserved_bins) ? l_ - reserved_bins : l_;

  const int bits = std::abs(cfg.baseband_input_bits);
  raw_bytes_ = n_ * (size_t)bits / 8;

  // boxcar ladder: 2, 4, ..., while <= max && < ts_count (the fused
  // ladder kernel handles up to 12 lengths = max boxcar 4096; exotic
  // configs beyond that are clamped rather than rejected)
  for (size_t L = 2;
       L <= cfg.max_boxcar_length && L < ts_count_ &&
       boxcar_lengths_.size() < 12;
       L *= 2)
    boxcar_lengths_.push_back(L);
  n_boxcars_ = (int)boxcar_lengths_.size();

  // SK corrected thresholds (reference rfi_mitigation.hpp:292-307)
  {
    const double M = (double)l_;
    double hi = cfg.sk_threshold, lo = 2.0 - hi;
    if (lo > hi) std::swap(lo, hi);
    const double corr = (M - 1.0) / (M + 1.0);
    sk_lo_ = (float)(lo * corr + 1.0);
    sk_hi_ = (float)(hi * corr + 1.0);
  }
  // normalization coefficient (reference rfi_mitigation_pipe.hpp:60-66)
  norm_coeff_ = (float)std::pow((double)nc_ * (double)nc_ / (double)s_, -0.5);

  f_min_ = cfg.freq_low;
  f_c_ = cfg.freq_low + cfg.bandwidth;
  df_ = cfg.bandwidth / (double)nc_;

  const int np = reduce_partials();
  for (int i = 0; i < n_slots_; ++i) slots_.emplace_back(new Slot());
  for (auto& sp : slots_) {
    Slot& s = *sp;
    check_hip(hipStreamCreateWithFlags(&s.stream, hipStreamNonBlocking),
              "stream create");
    check_hip(hipEventCreateWithFlags(&s.done, hipEventDisableTiming),
              "event create");
    check_hip(hipMalloc(&s.raw, raw_bytes_), "raw alloc");
    check_hip(hipMalloc(&s.samples, n_ * sizeof(float)), "samples alloc");
    check_hip(hipMalloc(&s.spec, (nc_ + 1) * sizeof(float2)), "spec alloc");
    check_hip(hipMalloc(&s.s2s4, s_ * sizeof(float2)), "s2s4 alloc");
    check_hip(hipMalloc(&s.flags, s_), "flags alloc");
    check_hip(hipMalloc(&s.ts, ts_count_ * sizeof(float)), "ts alloc");
    check_hip(hipMalloc(&s.ts_partial,
                        (size_t)time_series_chunks(ts_count_) * ts_count_ *
                            sizeof(float)),
              "ts partial alloc");
    check_hip(hipMalloc(&s.cumsum, ts_count_ * sizeof(float)), "cumsum alloc");
    check_hip(hipMalloc(&s.box, ts_count_ * sizeof(float)), "box alloc");
    check_hip(hipMalloc(&s.scan_scratch,
                        std::max(4096, scan_scratch_size(ts_count_)) *
                            sizeof(float)),
              "scan alloc");
    check_hip(hipMalloc(&s.partials, np * sizeof(double)), "partials alloc");
    if (n_boxcars_ > 0)
      check_hip(hipMalloc(&s.box_partials,
                          (size_t)n_boxcars_ * 2 * np * sizeof(double)),
                "box partials alloc");
    check_hip(hipMalloc(&s.mean_power, sizeof(double)), "mean alloc");
    check_hip(hipMalloc(&s.sums, 2 * sizeof(double)), "sums alloc");
    const int ncnt = 1 + 1 + n_boxcars_;  // zero_count + raw + boxcars
    check_hip(hipMalloc(&s.counters, ncnt * sizeof(unsigned)), "cnt alloc");
    check_hip(hipMalloc(&s.thresholds, (1 + n_boxcars_) * sizeof(float)),
              "thr alloc");
    check_hip(hipHostMalloc(&s.h_counters, ncnt * sizeof(unsigned)),
              "pinned cnt");
    check_hip(hipHostMalloc(&s.h_thresholds, (1 + n_boxcars_) * sizeof(float)),
              "pinned thr");
    // backend selection (see EngineConfig::fft_backend): the forward
    // 2^29-class packed C2C always favors the native plan; the batched
    // backward favors rocFFT below the measured l_ = 2^17 crossover
    fused_unpack_off_ = std::getenv("SRTB_NO_FUSED_UNPACK") != nullptr;
    const int be = cfg.fft_backend;
    native_fft_ = (be == 0 || be == 2) && NativeFft::supported(nc_);
    native_bwd_ = native_fft_ && NativeFft::supported(l_) &&
                  (be == 0 || l_ >= (1ull << 17));
    fuse_r2c_ = false;
    if (native_fft_) {
      s.nfwd.plan(nc_, 1, -1, s.stream);
      if (native_bwd_) {
        // the RFI+dedispersion preop fuses into the backward's FIRST
        // pass; SRTB_FFT_BWD32=1 plans it 32-max-column + final-256 (the
        // pair32 kernel runs the fp64 phase at 6 waves/SIMD).  A/B on one
        // box decides the default — kernel-sum favors pair32 (28.1 vs
        // 28.7 ms/block) but wall favored the wide plan on another box.
        const char* b32 = std::getenv("SRTB_FFT_BWD32");
        if (b32 && std::atoi(b32) != 0)
          s.nbwd.plan(l_, s_, +1, s.stream, /*maxcol_log2=*/5,
                      /*final_log2=*/8);
        else
          s.nbwd.plan(l_, s_, +1, s.stream);
        const int wpr = s.nbwd.dif_sk_wgs_per_row();
        if (cfg.enable_sk && wpr > 0)
          check_hip(hipMalloc(&s.sk_dif_partials,
                              s_ * (size_t)wpr * sizeof(float2)),
                    "sk dif partials");
      } else {
        s.plans.create_c2c_only(l_, s_, s.stream);
      }
      check_hip(hipStreamSynchronize(s.stream), "fft table sync");
      // r2c-into-backward fusion (opt-in SRTB_FUSE_R2C=1): the
      // pair-combine runs at the backward first pass's load and the RFI
      // mean comes from Parseval on the packed spectrum — the standalone
      // 8.6 GB r2c pass disappears.  MEASURED SLOWER end-to-end (50.8 vs
      // 61.5 Gsps, r02): the combine adds a second 4.3 GB read + sincos
      // to the pass that is already the chain's critical kernel (fp64
      // dedispersion), and the lost overlap outweighs the traffic saved.
      // Kept opt-in; numerics are exact (57 GPU tests pass either way).
      {
        const char* fe = std::getenv("SRTB_FUSE_R2C");
        const bool want = fe && std::atoi(fe) != 0;
        const int fwgs = s.nfwd.dif_sk_wgs_per_row();
        fuse_r2c_ = want && native_bwd_ && s.nbwd.first_pass_fusable() &&
                    fwgs > 0;
        if (fuse_r2c_) {
          fwd_pw_n_ = (size_t)fwgs;  // batch = 1
          check_hip(hipMalloc(&s.xbuf, nc_ * sizeof(float2)), "xbuf alloc");
          check_hip(hipMalloc(&s.fwd_pw, fwd_pw_n_ * sizeof(float2)),
                    "fwd pw alloc");
        }
      }
    } else {
      s.plans.create(n_, l_, s_, s.stream);
    }
  }

  if (cfg.window_kind != 0) {
    check_hip(hipMalloc(&window_, n_ * sizeof(float)), "window alloc");
    check_hip(build_window(window_, n_, cfg.window_kind, slots_[0]->stream),
              "window build");
    // K21: watfft window de-apply table of length l_ (reference
    // fft_pipe.hpp:350-358); only non-rectangle windows pay the pass
    check_hip(hipMalloc(&watfft_window_, l_ * sizeof(float)),
              "watfft window alloc");
    check_hip(build_window(watfft_window_, l_, cfg.window_kind,
                           slots_[0]->stream),
              "watfft window build");
    check_hip(hipStreamSynchronize(slots_[0]->stream), "window sync");
  }
  if (cfg.use_phase_table) {
    check_hip(hipMalloc(&phase_table_, nc_ * sizeof(float2)), "table alloc");
    check_hip(dedisp_phase_table(phase_table_, nc_, f_min_, f_c_, df_, cfg.dm,
                                 slots_[0]->stream),
              "phase table");
    check_hip(hipStreamSynchronize(slots_[0]->stream), "table sync");
  }
}

PipelineEngine::~PipelineEngine() {
  for (auto& sp : slots_) {
    if (sp->stream) (void)hipStreamSynchronize(sp->stream);
  }
  for (auto& sp : slots_) {
    Slot& s = *sp;
    s.plans.destroy();
    (void)hipFree(s.raw);
    (void)hipFree(s.samples);
    (void)hipFree(s.spec);
    (void)hipFree(s.s2s4);
    if (s.sk_dif_partials) (void)hipFree(s.sk_dif_partials);
    if (s.xbuf) (void)hipFree(s.xbuf);
    if (s.fwd_pw) (void)hipFree(s.fwd_pw);
    (void)hipFree(s.flags);
    (void)hipFree(s.ts);
    (void)hipFree(s.ts_partial);
    (void)hipFree(s.cumsum);
    (void)hipFree(s.box);
    (void)hipFree(s.scan_scratch);
    (void)hipFree(s.partials);
    if (s.box_partials) (void)hipFree(s.box_partials);
    (void)hipFree(s.mean_power);
    (void)hipFree(s.sums);
    (void)hipFree(s.counters);
    (void)hipFree(s.thresholds);
    (void)hipHostFree(s.h_counters);
    (void)hipHostFree(s.h_thresholds);
    if (s.graph_exec) (void)hipGraphExecDestroy(s.graph_exec);
    if (s.done) (void)hipEventDestroy(s.done);
    if (s.stream) (void)hipStreamDestroy(s.stream);
  }
  if (phase_table_) (void)hipFree(phase_table_);
  if (window_) (void)hipFree(window_);
  if (watfft_window_) (void)hipFree(watfft_window_);
}

void PipelineEngine::enqueue_chain(Slot& s, const uint8_t* dev_raw,
                                   const float* dev_samples, double dm,
                                   bool record_event) {
  hipStream_t st = s.stream;
  if (std::isnan(dm)) dm = cfg_.dm;
  const float2* table = phase_table_;
  if (table && dm != cfg_.dm)
    throw std::runtime_error("dm override requires use_phase_table=false");

  RoctxRange r_chain("srtb_block_chain");
  const float* fft_in = s.samples;
  // fused unpack: for the 2-bit rectangular-window native-forward case the
  // FFT's first column pass decodes the raw bytes itself (0.25 GB of byte
  // reads replace the unpack kernel's 4 GB write + the pass's 4 GB read)
  const int in_bits = cfg_.baseband_input_bits;
  const bool fuse_unpack = dev_raw && native_fft_ && !fused_unpack_off_ &&
                           (in_bits == 1 || in_bits == 2 || in_bits == 4 ||
                            in_bits == 8 || in_bits == -8 ||
                            in_bits == 16 || in_bits == -16) &&
                           !window_ && slots_[0]->nfwd.first_pass_fusable();
  if (dev_raw && !fuse_unpack) {
    // 1. unpack (+ window fused; default rectangle → none)
    check_hip(unpack(dev_raw, s.samples, n_, cfg_.baseband_input_bits,
                     window_, st),
              "unpack");
  } else if (!dev_raw && native_fft_) {
    // native fwd runs column passes in place on its input: copy the caller's
    // samples into the slot buffer first (D2D, overlapped on stream)
    check_hip(hipMemcpyAsync(s.samples, dev_samples, n_ * sizeof(float),
                             hipMemcpyDeviceToDevice, st),
              "samples d2d");
  } else if (!dev_raw) {
    fft_in = dev_samples;
  }
  const bool fr2c = fuse_r2c_;  // r2c pair-combine fused into bwd load
  if (native_fft_ && fr2c) {
    // 2. forward C2C of the packed-real view; the DIF store accumulates
    //    per-WG sum|Z|^2 and the RFI mean comes from Parseval — the
    //    standalone r2c pass (8.6 GB) is gone, its pair-combine runs at
    //    the backward first pass's load below.
    s.nfwd.exec(reinterpret_cast<float2*>(s.samples), s.spec, st,
                nullptr, cfg_.enable_rfi_s1 ? s.fwd_pw : nullptr,
                fuse_unpack ? dev_raw : nullptr, in_bits);
    if (cfg_.enable_rfi_s1)
      check_hip(r2c_mean_from_power(s.fwd_pw, fwd_pw_n_, s.spec, nc_,
                                    s.mean_power, st),
                "r2c mean");
  } else if (native_fft_) {
    // 2. forward C2C of the packed-real view + r2c post-process with FUSED
    //    mean-|X|^2 (saves the separate 4 GB mean_power pass)
    s.nfwd.exec(reinterpret_cast<float2*>(s.samples), s.spec, st,
                nullptr, nullptr, fuse_unpack ? dev_raw : nullptr, in_bits);
    check_hip(r2c_post_process(
                  s.spec, s.spec, nc_,
                  cfg_.enable_rfi_s1 ? s.partials : nullptr,
                  cfg_.enable_rfi_s1 ? s.mean_power : nullptr, st),
              "r2c post");
  } else {
    // 2. R2C forward (out-of-place; Nyquist bin written but ignored: the
    //    downstream count is Nc — reference drops it, fft_pipe.hpp:77)
    s.plans.exec_r2c(const_cast<float*>(fft_in), s.spec);
    // 3. mean |X|^2 over Nc
    if (cfg_.enable_rfi_s1)
      check_hip(mean_power(s.spec, nc_, s.partials, s.mean_power, st),
                "meanp");
  }
  // 4.+5. RFI s1 + manual zap + dedispersion fused into the backward FFT's
  // first column pass when the native planner allows it (saves a full
  // read+write of the 4 GB spectrum); otherwise the standalone fused kernel.
  float2* wf;
  const bool fuse_into_bwd = native_bwd_ && s.nbwd.first_pass_fusable();
  if (fuse_into_bwd) {
    FftPreop pre;
    pre.mean_power = cfg_.enable_rfi_s1 ? s.mean_power : nullptr;
    pre.threshold = cfg_.rfi_threshold;
    pre.norm_coeff = norm_coeff_;
    pre.n_zap = cfg_.n_zap_ranges;
    for (int i = 0; i < cfg_.n_zap_ranges; ++i) pre.zap[i] = cfg_.zap_ranges[i];
    pre.f_min = f_min_;
    pre.f_c = f_c_;
    pre.df = df_;
    pre.dm = dm;
    pre.table = table;
    pre.r2c_m = fr2c ? nc_ : 0;
    s.nbwd.exec(s.spec, reinterpret_cast<float2*>(s.samples), st, &pre,
                watfft_window_ ? nullptr : s.sk_dif_partials, nullptr, 2,
                fr2c ? s.xbuf : nullptr);
    wf = reinterpret_cast<float2*>(s.samples);
  } else {
    check_hip(rfi_dedisperse_fused(
                  s.spec, nc_, cfg_.enable_rfi_s1 ? s.mean_power : nullptr,
                  cfg_.rfi_threshold, norm_coeff_, cfg_.zap_ranges,
                  cfg_.n_zap_ranges, f_min_, f_c_, df_, dm, table, st),
              "rfi+dedisp");
    if (native_bwd_) {
      s.nbwd.exec(s.spec, reinterpret_cast<float2*>(s.samples), st);
      wf = reinterpret_cast<float2*>(s.samples);
    } else {
      s.plans.exec_c2c_backward(s.spec);
      wf = s.spec;
    }
  }
  s.wf = wf;
  if (watfft_window_)
    check_hip(window_deapply(wf, watfft_window_, s_ * l_, l_, st),
              "watfft deapply");

  const int ncnt = 2 + n_boxcars_;
  check_hip(hipMemsetAsync(s.counters, 0, ncnt * sizeof(unsigned), st),
            "memset counters");

  const uint8_t* ts_flags = nullptr;
  if (cfg_.enable_sk) {
    // 6. spectral kurtosis: row stats (fused into the backward DIF store
    //    when available — saves the 4 GB re-read) → flags → zap rows
    if (fuse_into_bwd && s.sk_dif_partials)
      check_hip(sk_combine_partials(s.sk_dif_partials, s_,
                                    s.nbwd.dif_sk_wgs_per_row(), s.s2s4, st),
                "sk combine");
    else
      check_hip(sk_row_stats(wf, s_, l_, s.s2s4, st), "sk stats");
    check_hip(sk_flags(wf, s.s2s4, s_, l_, sk_lo_, sk_hi_, s.flags,
                       s.counters + 0, st),
              "sk flags");
    // NOTE: the waterfall itself is NOT zapped here — the time-series pass
    // below applies the flags directly, and zeroing the flagged rows of the
    // 4 GB waterfall (~1 ms/block) only matters when the waterfall is
    // actually read (product dump on detection, display).  waterfall_ptr()
    // performs the zap on demand.
    ts_flags = s.flags;
  }
  // 7. time series over non-zapped rows
  check_hip(time_series_2stage(wf, ts_flags, s_, l_, ts_count_, s.ts,
                               s.ts_partial, st),
            "ts");
  // 8. baseline subtract
  check_hip(sum_sumsq(s.ts, ts_count_, s.partials, s.sums, st), "ts sum");
  check_hip(subtract_mean(s.ts, ts_count_, s.sums, st), "ts sub");
  // 9. raw-series detection
  check_hip(sum_sumsq(s.ts, ts_count_, s.partials, s.sums, st), "ts var");
  check_hip(count_above(s.ts, ts_count_, s.sums + 1, cfg_.snr_threshold,
                        s.counters + 1, s.thresholds + 0, st),
            "count raw");
  // 10. boxcar ladder from the inclusive scan — fused: thresholds and
  // counts for every length derive directly from the ~1 MB cumulative sum
  // in 3 launches (was 3 per length; the box series never materialize)
  if (n_boxcars_ > 0) {
    check_hip(inclusive_scan(s.ts, s.cumsum, ts_count_, s.scan_scratch, st),
              "scan");
    check_hip(boxcar_ladder(s.cumsum, ts_count_, boxcar_lengths_.data(),
                            n_boxcars_, s.box_partials, cfg_.snr_threshold,
                            s.thresholds + 1, s.counters + 2, st),
              "box ladder");
  }
  // 11. result counters → pinned host
  check_hip(hipMemcpyAsync(s.h_counters, s.counters, ncnt * sizeof(unsigned),
                           hipMemcpyDeviceToHost, st),
            "res d2h");
  check_hip(hipMemcpyAsync(s.h_thresholds, s.thresholds,
                           (1 + n_boxcars_) * sizeof(float),
                           hipMemcpyDeviceToHost, st),
            "thr d2h");
  if (record_event) {
    check_hip(hipEventRecord(s.done, st), "event record");
    s.busy = true;
  }
}

int PipelineEngine::submit(const void* host_bytes, size_t nbytes,
                           double dm_override) {
  if (nbytes != raw_bytes_) throw std::runtime_error("submit: wrong size");
  const int id = next_slot_;
  next_slot_ = (next_slot_ + 1) % n_slots_;
  Slot& s = *slots_[id];
  if (s.busy) {
    check_hip(hipEventSynchronize(s.done), "slot wait");
    s.busy = false;
  }
  const bool graph_ok = cfg_.use_hip_graph && std::isnan(dm_override);
  if (graph_ok && s.graph_exec && s.graph_host_src == host_bytes) {
    // steady state: replay the captured chain (H2D + kernels + D2H)
    check_hip(hipGraphLaunch(s.graph_exec, s.stream), "graph launch");
    check_hip(hipEventRecord(s.done, s.stream), "event record");
    s.busy = true;
    return id;
  }
  if (graph_ok && s.graph_pending >= 1 && !s.graph_exec) {
    // second submission from the same host buffer: capture it
    check_hip(hipStreamBeginCapture(s.stream, hipStreamCaptureModeThreadLocal),
              "begin capture");
    check_hip(hipMemcpyAsync(s.raw, host_bytes, nbytes,
                             hipMemcpyHostToDevice, s.stream),
              "raw h2d");
    enqueue_chain(s, s.raw, nullptr, dm_override, /*record_event=*/false);
    hipGraph_t g = nullptr;
    check_hip(hipStreamEndCapture(s.stream, &g), "end capture");
    check_hip(hipGraphInstantiate(&s.graph_exec, g, nullptr, nullptr, 0),
              "graph instantiate");
    check_hip(hipGraphDestroy(g), "graph destroy");
    s.graph_host_src = host_bytes;
    check_hip(hipGraphLaunch(s.graph_exec, s.stream), "graph launch");
    check_hip(hipEventRecord(s.done, s.stream), "event record");
    s.busy = true;
    return id;
  }
  if (graph_ok) {
    s.graph_pending = 1;
    s.graph_host_src = host_bytes;
  }
  check_hip(hipMemcpyAsync(s.raw, host_bytes, nbytes, hipMemcpyHostToDevice,
                           s.stream),
            "raw h2d");
  enqueue_chain(s, s.raw, nullptr, dm_override);
  return id;
}

int PipelineEngine::submit_device(const void* dev_bytes, size_t nbytes,
                                  double dm_override, hipEvent_t wait_event) {
  if (nbytes != raw_bytes_) throw std::runtime_error("submit: wrong size");
  const int id = next_slot_;
  next_slot_ = (next_slot_ + 1) % n_slots_;
  Slot& s = *slots_[id];
  if (s.busy) {
    check_hip(hipEventSynchronize(s.done), "slot wait");
    s.busy = false;
  }
  if (wait_event)
    check_hip(hipStreamWaitEvent(s.stream, wait_event, 0), "producer wait");
  enqueue_chain(s, static_cast<const uint8_t*>(dev_bytes), nullptr,
                dm_override);
  return id;
}

int PipelineEngine::submit_samples_device(const float* dev_samples,
                                          size_t count, double dm_override,
                                          hipEvent_t wait_event) {
  if (count != n_) throw std::runtime_error("submit_samples: wrong count");
  const int id = next_slot_;
  next_slot_ = (next_slot_ + 1) % n_slots_;
  Slot& s = *slots_[id];
  if (s.busy) {
    check_hip(hipEventSynchronize(s.done), "slot wait");
    s.busy = false;
  }
  if (wait_event)
    check_hip(hipStreamWaitEvent(s.stream, wait_event, 0), "fanout wait");
  enqueue_chain(s, nullptr, dev_samples, dm_override);
  return id;
}

BlockResult PipelineEngine::wait(int slot) {
  Slot& s = *slots_.at(slot);
  check_hip(hipEventSynchronize(s.done), "wait");
  s.busy = false;
  s.wf_zap_pending = cfg_.enable_sk;  // see waterfall_ptr()
  BlockResult r;
  r.zero_count = s.h_counters[0];
  r.counts.emplace_back(1u, s.h_counters[1]);
  for (int b = 0; b < n_boxcars_; ++b)
    r.counts.emplace_back((unsigned)boxcar_lengths_[b], s.h_counters[2 + b]);
  r.thresholds.assign(s.h_thresholds, s.h_thresholds + 1 + n_boxcars_);
  return r;
}

void PipelineEngine::synchronize() {
  for (auto& sp : slots_) {
    check_hip(hipStreamSynchronize(sp->stream), "sync");
    sp->busy = false;
  }
}

float2* PipelineEngine::waterfall_ptr(int slot) {
  Slot& s = *slots_.at(slot);
  float2* wf = s.wf ? s.wf : s.spec;
  if (s.wf_zap_pending) {
    // deferred SK row zap (flags are still resident for this slot); only
    // blocks whose waterfall is actually consumed pay the 4 GB pass
    check_hip(sk_zap_rows(wf, s.flags, s_, l_, s.stream), "sk zap lazy");
    check_hip(hipStreamSynchronize(s.stream), "sk zap sync");
    s.wf_zap_pending = false;
  }
  return wf;
}
float* PipelineEngine::time_series_ptr(int slot) { return slots_.at(slot)->ts; }
float* PipelineEngine::cumsum_ptr(int slot) { return slots_.at(slot)->cumsum; }
hipStream_t PipelineEngine::stream(int slot) { return slots_.at(slot)->stream; }

float* PipelineEngine::compute_boxcar(int slot, size_t L) {
  float* box = compute_boxcar_async(slot, L);
  check_hip(hipStreamSynchronize(slots_.at(slot)->stream), "boxcar sync");
  return box;
}

float* PipelineEngine::compute_boxcar_async(int slot, size_t L) {
  Slot& s = *slots_.at(slot);
  const size_t n_out = ts_count_ - L;
  check_hip(boxcar(s.cumsum, s.box, n_out, L, s.stream), "boxcar recompute");
  return s.box;
}

}  // namespace srtb_hip
