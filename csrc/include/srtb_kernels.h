// srtb_amd — MI355X-native kernel API (CDNA4 / gfx950, wave64).
//
// Raw-pointer + hipStream_t interface so the same kernels serve the torch
// extension, the native engine, and the standalone tools.  Each function
// enqueues asynchronously on the given stream and returns the first HIP error.
//
// Semantics mirror the reference pipeline (citations per kernel in the .hip
// files); the implementations are designed for gfx950: 64-wide wavefronts,
// vectorized (float4 / uint4) global access, grid-stride loops capped per
// CDNA guideline G11, two-pass deterministic reductions with fp64 partials.

#pragma once

#include <hip/hip_runtime.h>
#include <cstddef>
#include <cstdint>

namespace srtb_hip {

// ---------------- unpack (reference unpack.hpp:43-403) ----------------
// nbits: 1, 2, 4 = unsigned MSB-first packed; 8/-8, 16/-16, 32/-32 = u/int
// cast.  window (nullable) multiplies out[i] by window[i] (fused FFT window).
hipError_t unpack(const uint8_t* in, float* out, size_t out_count, int nbits,
                  const float* window, hipStream_t stream);

// int8 samples interleaved per-sample: p0 s0, p1 s0, p0 s1, ...
hipError_t unpack_interleaved_2pol(const int8_t* in, float* out0, float* out1,
                                   size_t count_per_pol, const float* window,
                                   hipStream_t stream);

// SNAP-1 "1 1 2 2": groups of 4 int8 = [s0p0 s1p0 s0p1 s1p1]
hipError_t unpack_naocpsr_snap1(const int8_t* in, float* out0, float* out1,
                                size_t count_per_pol, const float* window,
                                hipStream_t stream);

// GZNU A1: 4-byte words cycling over n_streams (2 or 4) ADCs; 4-stream
// variant is offset-binary (^0x80).  outs[s] may be null for unused streams.
hipError_t unpack_gznupsr_a1(const uint8_t* in, float* out0, float* out1,
                             float* out2, float* out3, int n_streams,
                             size_t count_per_stream, const float* window,
                             hipStream_t stream);

// FFT window coefficient table (0 = rectangle [no-op values], 1 = hann,
// 2 = hamming) — reference fft/fft_window.hpp.
hipError_t build_window(float* coef, size_t n, int kind, hipStream_t stream);

// ---------------- reductions ----------------
// Deterministic two-pass mean of |x|^2 over n complex bins.
// partials: device scratch of n_partials doubles (n_partials = reduce grid),
// out: 1 double = mean.  Use reduce_partials() to size the scratch.
int reduce_partials();  // number of fp64 partials the reductions use

hipError_t mean_power(const float2* in, size_t n, double* partials,
                      double* out_mean, hipStream_t stream);

// sum and sum-of-squares over a float array (two outputs in one pass):
// out[0] = sum, out[1] = sum of squares
hipError_t sum_sumsq(const float* in, size_t n, double* partials,
                     double* out2, hipStream_t stream);

// ---------------- RFI stage 1 + coherent dedispersion ----------------
// Standalone RFI s1: zap |x|^2 > threshold*mean, else scale by norm_coeff
// (reference rfi_mitigation_pipe.hpp:50-80).  mean is a device scalar.
hipError_t rfi_s1(float2* spec, size_t n, const double* mean_power,
                  float threshold, float norm_coeff, hipStream_t stream);

// Manual zap of inclusive bin range (reference rfi_mitigation.hpp:97-157).
hipError_t zap_bins(float2* spec, size_t lo, size_t hi, hipStream_t stream);

// Standalone coherent dedispersion, fp64 phase (reference
// coherent_dedispersion.hpp:133-248): x[i] *= exp(-2pi*i*frac(k)),
// k = D*1e6*dm/f*((f-f_c)/f_c)^2, f = f_min + df*i.
hipError_t dedisperse(float2* spec, size_t n, double f_min, double f_c,
                      double df, double dm, hipStream_t stream);

// Precompute the per-bin phase-factor table (table mode for fixed DM).
hipError_t dedisp_phase_table(float2* table, size_t n, double f_min,
                              double f_c, double df, double dm,
                              hipStream_t stream);

// Fused hot-path kernel: RFI-s1 zap + normalize + manual zap ranges +
// dedispersion phase rotation in ONE pass over the spectrum (the reference
// runs three separate kernels + waits; fusing removes two full HBM
// round-trips of the 4 GB spectrum).  zap_ranges: up to 16 inclusive [lo,hi]
// bin pairs, read from constant args.  factor_table: optional precomputed
// phases (else computed on the fly in fp64).
struct ZapRange { unsigned long long lo, hi; };
hipError_t rfi_dedisperse_fused(float2* spec, size_t n,
                                const double* mean_power, float threshold,
                                float norm_coeff, const ZapRange* ranges,
                                int n_ranges, double f_min, double f_c,
                                double df, double dm,
                                const float2* factor_table,
                                hipStream_t stream);

// ---------------- spectral kurtosis (stage 2) ----------------
// Per-row <sum |x|^2, sum |x|^4> of a [rows][len] waterfall; one workgroup
// per row (reference multi_mapreduce + rfi_mitigation.hpp:310-340).
hipError_t sk_row_stats(const float2* wf, size_t rows, size_t len,
                        float2* s2s4, hipStream_t stream);

// Decide zap flags from SK statistic: flag[i] = SK outside [lo_, hi_]
// (corrected thresholds precomputed on host), and count rows whose first
// sample is zero OR flagged (the reference's zapped-channel count observes
// column 0 after zapping).  zero_count must be zeroed beforehand.
hipError_t sk_flags(const float2* wf, const float2* s2s4, size_t rows,
                    size_t len, float lo_corrected, float hi_corrected,
                    uint8_t* flags, unsigned* zero_count, hipStream_t stream);

// Zero out flagged rows.
// wave-local small FFT: n in {256,512,1024}, one FFT per wave (batched,
// contiguous); tw_n = n-entry twiddle table with the sign baked in
hipError_t fft_wave_pass(const float2* in, float2* out, uint32_t n,
                         size_t n_ffts, int sign, const float2* tw_n,
                         hipStream_t stream);

// K21: waterfall window de-apply — wf[i] /= coef[i mod len] (reference
// fft_pipe.hpp:350-358; only for non-rectangle FFT windows)
hipError_t window_deapply(float2* wf, const float* coef, size_t total,
                          size_t len, hipStream_t stream);

hipError_t sk_zap_rows(float2* wf, const uint8_t* flags, size_t rows,
                       size_t len, hipStream_t stream);

// SK method 1 (reference rfi_mitigation.hpp:183-274): time-major [M][bins]
// waterfall; thresholds already corrected like sk_flags.  normalize applies
// the optional per-bin 1/sqrt(mean|x|^2) of surviving bins.
hipError_t sk_v1_stats(const float2* wf, size_t M, size_t bins, float2* s2s4,
                       hipStream_t stream);
hipError_t sk_v1_zap(float2* wf, size_t M, size_t bins, const float2* s2s4,
                     float lo_corrected, float hi_corrected, bool normalize,
                     hipStream_t stream);

// ---------------- signal detection ----------------
// ts[j] = sum_i |wf[i][j]|^2 over non-flagged rows, j < ts_count
// (reference signal_detect_pipe.hpp:305-316; flags may be null).
hipError_t time_series(const float2* wf, const uint8_t* flags, size_t rows,
                       size_t len, size_t ts_count, float* ts,
                       hipStream_t stream);

// Two-stage variant for large shapes: partial must hold
// time_series_chunks(ts_count) * ts_count floats; deterministic, ~10x the
// bandwidth of the single-pass kernel at S=2048, L=64k.
int time_series_chunks(size_t ts_count);
hipError_t time_series_2stage(const float2* wf, const uint8_t* flags,
                              size_t rows, size_t len, size_t ts_count,
                              float* ts, float* partial, hipStream_t stream);

// ts[i] -= sum/n (baseline subtract; sum is a device scalar from sum_sumsq).
hipError_t subtract_mean(float* ts, size_t n, const double* sum,
                         hipStream_t stream);

// threshold = snr * sqrt(sumsq/n) (device scalar sumsq of the zero-mean
// series); appends count of ts[i] > threshold into out_count (atomic; zero it
// first) and writes the threshold to out_threshold (reference
// signal_detect.hpp:33-67).
hipError_t count_above(const float* ts, size_t n, const double* sumsq,
                       float snr, unsigned* out_count, float* out_threshold,
                       hipStream_t stream);

// Inclusive prefix sum (float in → float out); scratch must hold
// scan_scratch_size(n) floats (one per 2048-element block).
int scan_scratch_size(size_t n);
hipError_t inclusive_scan(const float* in, float* out, size_t n,
                          float* scratch, hipStream_t stream);

// box[i] = cumsum[i+L] - cumsum[i], i < n_out (reference
// signal_detect_pipe.hpp:387-423).
// fused boxcar ladder: thresholds + counts for up to 12 lengths in 3
// launches, derived directly from the cumulative sum (no box arrays).
// partials: device scratch of n_lengths * 2 * reduce_partials() doubles;
// out_thr/out_counts: [n_lengths] each (counts must be zeroed beforehand).
hipError_t boxcar_ladder(const float* cumsum, size_t ts_count,
                         const size_t* lengths, int n_lengths,
                         double* partials, float snr, float* out_thr,
                         unsigned* out_counts, hipStream_t stream);

hipError_t boxcar(const float* cumsum, float* out, size_t n_out, size_t L,
                  hipStream_t stream);

// ---------------- hand-written Stockham FFT ----------------
// One generic pass of the LDS-staged Stockham FFT (csrc/kernels/fft.hip).
// Index-math oracle: srtb_amd/fftref.py.  Host-side planning:
// csrc/fft/native_fft.h.
struct FftPassDesc {
  uint32_t n;            // pow2 FFT length of this pass (<= 4096)
  uint32_t d0 = 1, d1 = 1;  // instance id -> digits q0, q1, q2 (pow2;
                            // d0 == 0 means "q0 = id" for contiguous rows)
  unsigned long long in_c0 = 0, in_c1 = 0, in_c2 = 0;
  unsigned long long in_stride = 1;
  unsigned long long out_c0 = 0, out_c1 = 0, out_c2 = 0;
  unsigned long long out_stride = 1;
  unsigned long long tw_f0 = 0, tw_f1 = 0;
  unsigned long long tw_mod = 0;  // 0 = no inter-pass twiddle
  int tw_lo_bits = 0;
  double tw_angle = 0.0;  // sign * 2*pi / tw_mod (for computed twiddles)
};

// build table[j] = exp(sign * 2*pi*i * j / m), j in [0, count)
hipError_t fft_build_twiddle(float2* table, size_t count, double m, int sign,
                             hipStream_t stream);

// tw_n: FULL-circle per-length table (n entries, staged into LDS)
hipError_t fft_stockham_pass(const float2* in, float2* out,
                             const FftPassDesc& d, size_t n_ffts, int F,
                             bool load_ffast, bool store_ffast, int sign,
                             const float2* tw_n, const float2* tw_hi,
                             const float2* tw_lo, hipStream_t stream);

// Optional per-element pre-op fused into a column pass's LOADS: the full
// RFI-s1 zap + normalize + manual zap + coherent dedispersion of
// rfi_dedisperse_fused, applied at the element's flat spectrum index
// (= its load address offset).  Fusing it into the backward FFT's first
// pass saves a whole read+write of the 4 GB spectrum per block.
struct FftPreop {
  const double* mean_power = nullptr;  // null = no RFI zap/normalize
  float threshold = 0.f, norm_coeff = 1.f;
  int n_zap = 0;
  ZapRange zap[16] = {};
  double f_min = 0, f_c = 0, df = 0, dm = 0;
  // optional cached dedispersion factors [nc]; when set the pass reads the
  // table (one extra coalesced float2 load) instead of the fp64 phase
  // computation, which would otherwise double the pass's VALU cost
  const float2* table = nullptr;
  // != 0: the pass input is the PACKED forward spectrum Z (length r2c_m);
  // the r2c pair-combine X[k] = E + w_k O runs at load (eliminates the
  // standalone r2c pass; needs an out-of-place first pass)
  unsigned long long r2c_m = 0;
};

// mean |X|^2 over the r2c spectrum from the PACKED spectrum Z via
// Parseval: sum_{k<m}|X_k|^2 = sum|Z|^2 + (X0^2 - XM^2)/2 with
// X0 = Re Z0 + Im Z0, XM = Re Z0 - Im Z0.  power_partials: per-workgroup
// <sum|Z|^2, _> pairs accumulated by the forward DIF store (the SK-fusion
// machinery), z0 = device pointer to Z[0].
hipError_t r2c_mean_from_power(const float2* power_partials,
                               size_t n_partials, const float2* z0, size_t m,
                               double* out_mean, hipStream_t stream);

// Register-resident column FFT pass (N in {2,4,8,16,32,64}; one FFT per
// thread fully in VGPRs; in-place: out must alias layout of in addressing;
// uses the same FftPassDesc fields; out_* ignored, stores to input layout).
// raw2 (optional): fused sub-byte unpack (raw_bits in {1,2,4}) — the pass
// loads and decodes raw bytes instead of reading `in` (forward first pass
// only; requires an inter-pass twiddle and no preop).
hipError_t fft_col_pass(const float2* in, float2* out, const FftPassDesc& d,
                        size_t n_ffts, int sign, const float2* tw_n,
                        const float2* tw_hi, const float2* tw_lo,
                        hipStream_t stream, const FftPreop* preop = nullptr,
                        const uint8_t* raw2 = nullptr, int raw_bits = 2);

// Final composite pass: in-place radix-4 DIF in LDS with base-4
// digit-reversal folded into the store, plus multi-digit output scatter.
// Input must be fully linear (instance blocks contiguous).  n = 4^t.
struct DifFinalDesc {
  uint32_t n;
  unsigned long long out_c2;         // batch-row coefficient
  unsigned long long out_elem_coef;  // output element stride
  int n_prefix = 0;                  // prefix digits, extraction order
  int pf_bits[4] = {0, 0, 0, 0};
  unsigned long long pf_coef[4] = {0, 0, 0, 0};
};
// sk_partials (optional): DifFinalDesc rows own whole batch rows when
// (n_ffts/batch) % F == 0; each workgroup then accumulates <sum|x|^2,
// sum|x|^4> of its stored elements into
// sk_partials[row * wgs_per_row + wg_in_row] (deterministic, no atomics) —
// this replaces the separate 4 GB sk_row_stats read.  wgs_per_row =
// (L/n)/F.
// EXPERIMENTAL strided 512-point LDS middle pass (see fft.hip); reachable
// only via the SRTB_FFT_FACTORS plan override.
hipError_t fft_mid512_pass(const float2* in, float2* out,
                           const FftPassDesc& d, size_t n_ffts, int sign,
                           const float2* tw_n, hipStream_t stream);

hipError_t fft_dif_final(const float2* in, float2* out, const DifFinalDesc& d,
                         size_t n_ffts, int F, int sign, const float2* tw_n,
                         float2* sk_partials, hipStream_t stream);

// combine the DIF-written partials into per-row <S2,S4>
hipError_t sk_combine_partials(const float2* partials, size_t rows,
                               int wgs_per_row, float2* s2s4,
                               hipStream_t stream);

// packed-real R2C finish: Z = C2C(x_even + i*x_odd) of length m -> true
// spectrum X[0..m) (Nyquist dropped).  In-place safe (x may alias z).
// If mean_partials != null (>= 1024 doubles), also writes mean|X|^2 to
// out_mean (fused RFI-s1 statistic).
hipError_t r2c_post_process(const float2* z, float2* x, size_t m,
                            double* mean_partials, double* out_mean,
                            hipStream_t stream);

// ---------------- display / spectrum simplification ----------------
// Area-averaged power resample of [rows][len] complex waterfall to [H][W]
// floats; one workgroup (64 lanes) per output pixel with LDS tree reduce
// (reference resample_spectrum_3, simplify_spectrum.hpp:423-620; wg=64 was
// measured fastest on wave64 GCN — kept for CDNA4).
hipError_t resample_power_2d(const float2* wf, size_t rows, size_t len,
                             float* out, int out_h, int out_w,
                             hipStream_t stream);

// sum over img (H*W floats) into partials/out via sum_sumsq; then
// img[i] *= 1/(2*mean) with mean = sum/n (reference simplify_spectrum.hpp:627-644).
hipError_t normalize_by_mean(float* img, size_t n, const double* sum,
                             hipStream_t stream);

// intensity [0,1] → ARGB32 lerp color_0..color_1, else overflow color
// (reference simplify_spectrum.hpp:700-731, config.hpp:60-68).
hipError_t generate_pixmap(const float* intensity, uint32_t* out, size_t n,
                           uint32_t color0, uint32_t color1,
                           uint32_t color_overflow, hipStream_t stream);

// ---------------- misc device algorithms ----------------
// Running-mean 1-bit threshold, time-major [nsamp][nchan]
// (reference algorithm/running_mean.hpp:31-77).
hipError_t running_mean_init(const float* data, size_t nsamp, size_t nchan,
                             size_t windowsize, float* ave, hipStream_t stream);
hipError_t running_mean(const float* data, size_t nsamp, size_t nchan,
                        uint8_t* out, size_t windowsize, float* ave,
                        hipStream_t stream);

// mag[i] = |x[i]| for complex input (correlator backward-FFT epilogue)
hipError_t complex_abs(const float2* x, float* mag, size_t n,
                       hipStream_t stream);

// correlator pointwise: corr[i] = scale * f1[i]*conj(f2[i]); mag[i] = |corr[i]|
// (reference src/correlator.cpp:116-140; mag may be null).
hipError_t correlate_pointwise(const float2* f1, const float2* f2,
                               float2* corr, float* mag, size_t n, float scale,
                               hipStream_t stream);

}  // namespace srtb_hip
