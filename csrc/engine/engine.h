// PipelineEngine — the MI355X-native per-GPU streaming engine.
//
// Runs the full per-block chain of the reference's streaming data path
//  with zero host synchronization inside a block:
//
//   H2D(raw) → unpack(+window) → R2C FFT → mean-power reduce →
//   fused {RFI-s1 zap + normalize + manual zap + coherent dedispersion} →
//   batched backward C2C (waterfall) → SK row stats → flags → zap rows →
//   time series → baseline subtract → threshold counts → prefix scan →
//   boxcar ladder (2,4,…,max) with per-length thresholds →
//   D2H(result counters) → event
//
// Two slots, each with its own HIP stream, device buffers and hipFFT plans:
// while slot A computes, slot B's next block uploads (the reference instead
// waits after every kernel — SURVEY.md §2b note).  Results (detection
// counters) land in pinned memory; wait(slot) is the only host sync.

#pragma once

#include <hip/hip_runtime.h>

#include <array>
#include <cstdint>
#include <memory>
#include <vector>

#include "../fft/fft_plans.h"
#include "../fft/native_fft.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

struct EngineConfig {
  size_t baseband_input_count = 1ull << 28;  // N real samples per block
  int baseband_input_bits = 8;               // 1/2/4/8/-8/16/-16/32/-32
  size_t spectrum_channel_count = 1ull << 15;  // S
  double freq_low = 1000.0;    // MHz
  double bandwidth = 500.0;    // MHz (may be negative)
  double sample_rate = 1e9;    // samples/s
  double dm = 0.0;
  float rfi_threshold = 10.0f;          // mitigate_rfi_average_method_threshold
  float sk_threshold = 1.1f;            // mitigate_rfi_spectral_kurtosis_threshold
  float snr_threshold = 6.0f;           // signal_detect_signal_noise_threshold
  size_t max_boxcar_length = 1024;
  size_t nsamps_reserved = 0;           // overlap (real samples), from host
  int n_zap_ranges = 0;
  ZapRange zap_ranges[16] = {};
  bool use_phase_table = false;  // cache dedispersion factors for fixed DM
  bool enable_rfi_s1 = true;
  bool enable_sk = true;
  // 0 = native always, 1 = hipFFT always, 2 = auto (native forward; the
  // batched backward picks rocFFT below the measured 2^17 crossover —
  // rocFFT's specialized sbcc/sbrc kernels win 1.4-2.9x for lengths
  // ≤ 2^16, the native column+DIF plan wins 2.2x at 2^18)
  int fft_backend = 2;
  // FFT window fused into unpack: 0 = rectangle (reference default),
  // 1 = hann, 2 = hamming
  int window_kind = 0;
  // capture the steady-state per-block chain (~55 launches) into a hipGraph
  // per slot and replay it; falls back to direct enqueue whenever the
  // submission differs from the captured one (dm override, other host ptr)
  bool use_hip_graph = false;
};

struct BlockResult {
  unsigned zero_count = 0;  // zapped-channel count (gate detections on host)
  // (boxcar_length, signal_count) — boxcar_length 1 is the raw series
  std::vector<std::pair<unsigned, unsigned>> counts;
  std::vector<float> thresholds;
};

class PipelineEngine {
 public:
  explicit PipelineEngine(const EngineConfig& cfg, int n_slots = 2);
  ~PipelineEngine();
  PipelineEngine(const PipelineEngine&) = delete;
  PipelineEngine& operator=(const PipelineEngine&) = delete;

  // Upload raw bytes (host, ideally pinned) and enqueue the whole chain on
  // the slot's stream.  Blocks only if the slot's previous block is still in
  // flight.  Returns the slot used.
  // dm_override: NAN = use the configured DM; a finite value re-runs the
  // (on-the-fly fp64) dedispersion at that DM — the DM-trial sweep path.
  int submit(const void* host_bytes, size_t nbytes, double dm_override = NAN);

  // Enqueue the chain reading raw bytes already on the device (e.g. from a
  // torch tensor); caller guarantees lifetime until wait().  wait_event
  // (optional): slot stream waits on it first — pass an event recorded on
  // the stream that produced dev_bytes (e.g. torch's current stream).
  int submit_device(const void* dev_bytes, size_t nbytes,
                    double dm_override = NAN, hipEvent_t wait_event = nullptr);

  // Enqueue the chain starting at the R2C FFT from already-unpacked float
  // samples on the device (count = baseband_input_count).  Used for
  // multi-polarization formats whose unpack fans one packet stream out into
  // several sample streams (reference unpack_pipe.hpp:146-390).
  // wait_event (optional): the slot's stream waits on it before the chain —
  // lets an app-owned fan-out stream (H2D + unpack of the packed block)
  // feed several engines without host synchronization.
  int submit_samples_device(const float* dev_samples, size_t count,
                            double dm_override = NAN,
                            hipEvent_t wait_event = nullptr);

  // Wait for a slot's chain and return its detection counters.
  BlockResult wait(int slot);

  // Wait for all outstanding work.
  void synchronize();

  // Geometry
  size_t n() const { return n_; }
  size_t nc() const { return nc_; }
  size_t n_channels() const { return s_; }
  size_t waterfall_len() const { return l_; }
  size_t ts_count() const { return ts_count_; }
  size_t raw_bytes() const { return raw_bytes_; }
  int n_boxcars() const { return n_boxcars_; }
  int n_slots() const { return n_slots_; }

  // Device pointers of a slot (valid after wait(slot)) — for tests, dumps and
  // the display path.
  float2* waterfall_ptr(int slot);   // [S][L]
  float* time_series_ptr(int slot);  // [ts_count]
  float* cumsum_ptr(int slot);       // [ts_count]
  // Recompute one boxcar series into the slot's boxcar buffer and return it
  // (for dumping a detected series).
  float* compute_boxcar(int slot, size_t boxcar_length);
  // Async variant: enqueue the recompute on the slot's stream and return the
  // device pointer WITHOUT synchronizing (follow with hipMemcpyAsync on the
  // same stream + an event for a fully async detection dump).
  float* compute_boxcar_async(int slot, size_t boxcar_length);

  hipStream_t stream(int slot);

 private:
  struct Slot;
  void enqueue_chain(Slot& s, const uint8_t* dev_raw,
                     const float* dev_samples, double dm,
                     bool record_event = true);

  EngineConfig cfg_;
  size_t n_, nc_, s_, l_, ts_count_, raw_bytes_;
  int n_boxcars_ = 0;
  std::vector<size_t> boxcar_lengths_;
  float sk_lo_ = 0, sk_hi_ = 0;
  float norm_coeff_ = 1.0f;
  double f_min_ = 0, f_c_ = 0, df_ = 0;
  int n_slots_ = 2;

  float2* phase_table_ = nullptr;  // shared across slots (read-only)
  float* window_ = nullptr;        // fused FFT window table (null = rect)
  float* watfft_window_ = nullptr;  // K21 de-apply table, length l_
  bool native_fft_ = false;        // hand-written FORWARD FFT active
  bool fused_unpack_off_ = false;  // SRTB_NO_FUSED_UNPACK kill switch
  bool native_bwd_ = false;        // hand-written BACKWARD (waterfall) FFT
  bool fuse_r2c_ = false;          // r2c pair-combine fused into bwd load
  size_t fwd_pw_n_ = 0;            // fwd power partial count


  struct Slot {
    hipStream_t stream = nullptr;
    hipEvent_t done = nullptr;
    bool busy = false;
    bool wf_zap_pending = false;    // SK zap deferred until waterfall read
    float2* wf = nullptr;           // where the waterfall landed last block
    uint8_t* raw = nullptr;         // device raw bytes
    float* samples = nullptr;       // [N] unpacked
    float2* spec = nullptr;         // [Nc+1] spectrum / waterfall (in-place)
    float2* xbuf = nullptr;         // [Nc] bwd working set (r2c fusion)
    float2* fwd_pw = nullptr;       // fwd-DIF per-WG power partials
    float2* s2s4 = nullptr;         // [S]
    float2* sk_dif_partials = nullptr;  // [S * wgs_per_row] (fused SK stats)
    uint8_t* flags = nullptr;       // [S]
    float* ts = nullptr;            // [ts_count]
    float* ts_partial = nullptr;    // [chunks][ts_count] two-stage scratch
    float* cumsum = nullptr;        // [ts_count]
    float* box = nullptr;           // [ts_count]
    float* scan_scratch = nullptr;  // [4096]
    double* partials = nullptr;     // reduce partials + out scalars
    double* box_partials = nullptr; // [n_boxcars][2][reduce_partials]
    double* mean_power = nullptr;   // -> inside partials block
    double* sums = nullptr;         // [2]
    unsigned* counters = nullptr;   // [1 + n_boxcars + 1] zero_count + counts
    float* thresholds = nullptr;    // [1 + n_boxcars]
    unsigned* h_counters = nullptr; // pinned result mirror
    float* h_thresholds = nullptr;  // pinned
    FftPlanSet plans;               // hipFFT fallback
    NativeFft nfwd, nbwd;           // hand-written path
    // hipGraph replay state
    hipGraphExec_t graph_exec = nullptr;
    const void* graph_host_src = nullptr;  // host ptr the capture used
    int graph_pending = 0;  // submissions seen for this slot
  };
  std::vector<std::unique_ptr<Slot>> slots_;
  int next_slot_ = 0;
};

}  // namespace srtb_hip
