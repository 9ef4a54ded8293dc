// Spectrum-domain kernels: reductions, RFI stage 1, coherent dedispersion
// (fused hot path), spectral kurtosis, time-series detection, scan, boxcar.
//
// Reference semantics: rfi_mitigation_pipe.hpp:50-101, rfi_mitigation.hpp,
// coherent_dedispersion.hpp:133-248, signal_detect_pipe.hpp:252-441,
// signal_detect.hpp:25-70, algorithm/{map_reduce,multi_reduce}.hpp.
//
// MI355X design notes:
//  - all reductions are two-pass and deterministic (fixed partial count,
//    fp64 partials): pass 1 = grid-stride block sums, pass 2 = one block.
//  - the RFI-s1 + manual-zap + dedispersion chain is ONE fused kernel: the
//    reference walks the 4 GB spectrum three times + waits between kernels;
//    here it is read once / written once (float4 vector I/O), with the fp64
//    phase computed in-kernel (modf + sincos on the wrapped argument) or read
//    from a precomputed table for fixed-DM streaming.
//  - spectral kurtosis: one 256-lane workgroup per frequency row, strided
//    float4 loads, wave64 shuffle + LDS tree reduce.

#include "common.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

namespace {

constexpr int kReducePartials = 1024;

// ---------------- two-pass reductions ----------------

__global__ void k_mean_power_pass1(const float2* __restrict__ in, size_t n,
                                   double* __restrict__ partials) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  // vectorized main loop: float4 = 2 complex bins
  const size_t n2 = n / 2;
  const float4* in4 = reinterpret_cast<const float4*>(in);
  float acc = 0.0f;
  double accd = 0.0;
  size_t iters = 0;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n2;
       i += stride) {
    const float4 v = in4[i];
    acc += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    // drain to fp64 periodically so huge blocks don't lose precision
    if (((++iters) & 1023) == 0) { accd += acc; acc = 0.0f; }
  }
  accd += acc;
  // odd tail (n is even in practice; keep correct anyway)
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    const float2 v = in[n - 1];
    accd += (double)v.x * v.x + (double)v.y * v.y;
  }
  const double b = block_reduce_sum(accd);
  if (threadIdx.x == 0) partials[blockIdx.x] = b;
}

__global__ void k_reduce_partials_mean(const double* __restrict__ partials,
                                       int np, size_t n,
                                       double* __restrict__ out_mean) {
  double acc = 0.0;
  for (int i = threadIdx.x; i < np; i += blockDim.x) acc += partials[i];
  const double b = block_reduce_sum(acc);
  if (threadIdx.x == 0) *out_mean = b / (double)n;
}

__global__ void k_sum_sumsq_pass1(const float* __restrict__ in, size_t n,
                                  double* __restrict__ partials) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  double s = 0.0, s2 = 0.0;
  const size_t n4 = n / 4;
  const float4* in4 = reinterpret_cast<const float4*>(in);
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    const float4 v = in4[i];
    s += (double)v.x + v.y + v.z + v.w;
    s2 += (double)v.x * v.x + (double)v.y * v.y + (double)v.z * v.z +
          (double)v.w * v.w;
  }
  if (blockIdx.x == 0 && threadIdx.x < (n & 3)) {
    const float v = in[4 * n4 + threadIdx.x];
    s += v;
    s2 += (double)v * v;
  }
  const double bs = block_reduce_sum(s);
  const double bs2 = block_reduce_sum(s2);
  if (threadIdx.x == 0) {
    partials[blockIdx.x] = bs;
    partials[kReducePartials + blockIdx.x] = bs2;
  }
}

__global__ void k_reduce_partials_2(const double* __restrict__ partials,
                                    int np, double* __restrict__ out2) {
  double s = 0.0, s2 = 0.0;
  for (int i = threadIdx.x; i < np; i += blockDim.x) {
    s += partials[i];
    s2 += partials[kReducePartials + i];
  }
  const double bs = block_reduce_sum(s);
  const double bs2 = block_reduce_sum(s2);
  if (threadIdx.x == 0) {
    out2[0] = bs;
    out2[1] = bs2;
  }
}

// ---------------- RFI s1 + dedispersion ----------------

__device__ inline float2 dedisp_factor(size_t i, double f_min, double f_c,
                                       double df, double dm) {
  return srtb_dedisp_factor(i, f_min, f_c, df, dm);
}

__device__ inline float2 cmul(float2 a, float2 b) {
  return make_float2(a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x);
}

struct ZapRangesArg {
  ZapRange r[16];
  int count;
};

template <bool kTable, bool kRfi>
__global__ void k_rfi_dedisp_fused(float2* __restrict__ spec, size_t n,
                                   const double* __restrict__ mean_power,
                                   float threshold, float norm_coeff,
                                   ZapRangesArg zaps, double f_min, double f_c,
                                   double df, double dm,
                                   const float2* __restrict__ table) {
  const float thr_mean = kRfi ? threshold * (float)(*mean_power) : 0.0f;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float2 v = spec[i];
    bool zap = false;
    if constexpr (kRfi) {
      zap = norm2(v) > thr_mean;
    }
    for (int z = 0; z < zaps.count; ++z)
      zap |= (i >= zaps.r[z].lo) & (i <= zaps.r[z].hi);
    if (zap) {
      spec[i] = make_float2(0.0f, 0.0f);
    } else {
      if constexpr (kRfi) {
        v.x *= norm_coeff;
        v.y *= norm_coeff;
      }
      const float2 fac =
          kTable ? table[i] : dedisp_factor(i, f_min, f_c, df, dm);
      spec[i] = cmul(v, fac);
    }
  }
}

__global__ void k_rfi_s1(float2* __restrict__ spec, size_t n,
                         const double* __restrict__ mean_power,
                         float threshold, float norm_coeff) {
  const float thr_mean = threshold * (float)(*mean_power);
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float2 v = spec[i];
    if (norm2(v) > thr_mean)
      spec[i] = make_float2(0.0f, 0.0f);
    else
      spec[i] = make_float2(v.x * norm_coeff, v.y * norm_coeff);
  }
}

__global__ void k_zap_bins(float2* __restrict__ spec, size_t lo, size_t count) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < count;
       i += stride)
    spec[lo + i] = make_float2(0.0f, 0.0f);
}

__global__ void k_dedisperse(float2* __restrict__ spec, size_t n, double f_min,
                             double f_c, double df, double dm) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    spec[i] = cmul(spec[i], dedisp_factor(i, f_min, f_c, df, dm));
}

__global__ void k_phase_table(float2* __restrict__ table, size_t n,
                              double f_min, double f_c, double df, double dm) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    table[i] = dedisp_factor(i, f_min, f_c, df, dm);
}

// ---------------- spectral kurtosis ----------------

// one workgroup per frequency row; strided float4 loads over len complex
__global__ void k_sk_row_stats(const float2* __restrict__ wf, size_t len,
                               float2* __restrict__ s2s4) {
  const size_t row = blockIdx.x;
  const float2* rp = wf + row * len;
  const float4* rp4 = reinterpret_cast<const float4*>(rp);
  const size_t len2 = len / 2;
  float s2 = 0.0f, s4 = 0.0f;
  for (size_t i = threadIdx.x; i < len2; i += blockDim.x) {
    const float4 v = rp4[i];
    const float p0 = v.x * v.x + v.y * v.y;
    const float p1 = v.z * v.z + v.w * v.w;
    s2 += p0 + p1;
    s4 += p0 * p0 + p1 * p1;
  }
  if (threadIdx.x == 0 && (len & 1)) {
    const float p = norm2(rp[len - 1]);
    s2 += p;
    s4 += p * p;
  }
  const float bs2 = block_reduce_sum(s2);
  const float bs4 = block_reduce_sum(s4);
  if (threadIdx.x == 0) s2s4[row] = make_float2(bs2, bs4);
}

// scalar fallback when rows are not 16-B aligned (odd len)
__global__ void k_sk_row_stats_scalar(const float2* __restrict__ wf,
                                      size_t len, float2* __restrict__ s2s4) {
  const size_t row = blockIdx.x;
  const float2* rp = wf + row * len;
  float s2 = 0.0f, s4 = 0.0f;
  for (size_t i = threadIdx.x; i < len; i += blockDim.x) {
    const float p = norm2(rp[i]);
    s2 += p;
    s4 += p * p;
  }
  const float bs2 = block_reduce_sum(s2);
  const float bs4 = block_reduce_sum(s4);
  if (threadIdx.x == 0) s2s4[row] = make_float2(bs2, bs4);
}

__global__ void k_sk_flags(const float2* __restrict__ wf,
                           const float2* __restrict__ s2s4, size_t rows,
                           size_t len, float lo_, float hi_,
                           uint8_t* __restrict__ flags,
                           unsigned* __restrict__ zero_count) {
  const size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= rows) return;
  const float2 p = s2s4[i];
  const float sk = (float)len * (p.y / (p.x * p.x));
  const bool zap = (sk > hi_) || (sk < lo_) || !(p.x > 0.0f);
  flags[i] = zap ? 1 : 0;
  // reference counts channels whose first time sample is zero AFTER zapping
  // (signal_detect_pipe.hpp:261-281)
  const bool first_zero = zap || (norm2(wf[i * len]) == 0.0f);
  if (first_zero) atomicAdd(zero_count, 1u);
}

// 2-D grid: y = row (whole workgroup early-outs on unflagged rows — the
// 1-D version spent 86% VALU on a per-element 64-bit division)
__global__ void k_sk_zap_rows(float2* __restrict__ wf,
                              const uint8_t* __restrict__ flags, size_t rows,
                              size_t len) {
  const size_t row = blockIdx.y;
  if (!flags[row]) return;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  float2* rp = wf + row * len;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < len;
       i += stride)
    rp[i] = make_float2(0.0f, 0.0f);
}

// SK method 1 (reference rfi_mitigation.hpp:183-274): TIME-MAJOR layout
// [M][bins]; one thread per frequency bin walks its column (stride = bins,
// coalesced across threads), then a zero/normalize pass.
__global__ void k_sk_v1_stats(const float2* __restrict__ wf, size_t M,
                              size_t bins, float2* __restrict__ s2s4) {
  const size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= bins) return;
  float s2 = 0.0f, s4 = 0.0f;
  for (size_t m = 0; m < M; ++m) {
    const float p = norm2(wf[m * bins + j]);
    s2 += p;
    s4 += p * p;
  }
  s2s4[j] = make_float2(s2, s4);
}

template <bool kNormalize>
__global__ void k_sk_v1_zap(float2* __restrict__ wf, size_t M, size_t bins,
                            const float2* __restrict__ s2s4, float lo_,
                            float hi_) {
  const size_t total = M * bins;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const size_t j = i % bins;  // frequency bin (bins need not be pow2 here)
    const float2 p = s2s4[j];
    const float sk = (float)M * (p.y / (p.x * p.x));
    const bool zap = (sk > hi_) || (sk < lo_) || !(p.x > 0.0f);
    if (zap) {
      wf[i] = make_float2(0.0f, 0.0f);
    } else if constexpr (kNormalize) {
      const float scale = rsqrtf(p.x / (float)M);
      wf[i] = make_float2(wf[i].x * scale, wf[i].y * scale);
    }
  }
}

// ---------------- time series + detection ----------------

// ts[j] = sum over non-flagged rows of |wf[row][j]|^2.
//
// Two-stage for MI355X: a single thread-per-column kernel over S=2048 rows
// launches only ts_count/256 workgroups (≈1 per CU at L=64k) and each thread
// chases 512 KB-strided dependent loads — measured 0.5 TB/s.  Stage 1 splits
// the rows into chunks (grid = col_blocks × chunks, each block sums its chunk
// into partial[chunk][j] with float4 column-pair loads); stage 2 reduces the
// chunks.  Deterministic (no atomics).
template <bool kFlags>
__global__ void k_time_series_partial(const float2* __restrict__ wf,
                                      const uint8_t* __restrict__ flags,
                                      size_t rows, size_t len, size_t ts_count,
                                      float* __restrict__ partial,
                                      int n_chunks) {
  const int chunk = blockIdx.y;
  const size_t rows_per = (rows + n_chunks - 1) / n_chunks;
  const size_t r0 = (size_t)chunk * rows_per;
  const size_t r1 = min(r0 + rows_per, rows);
  // each thread owns two adjacent columns (one float4 load per row)
  const size_t j2 = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) * 2;
  if (j2 >= ts_count) return;
  const bool pair = (j2 + 1 < ts_count);
  float acc0 = 0.0f, acc1 = 0.0f;
#pragma unroll 4
  for (size_t i = r0; i < r1; ++i) {
    if (kFlags && flags[i]) continue;
    const float2* rp = wf + i * len + j2;
    if (pair) {
      const float4 v = *reinterpret_cast<const float4*>(rp);
      acc0 += v.x * v.x + v.y * v.y;
      acc1 += v.z * v.z + v.w * v.w;
    } else {
      acc0 += norm2(*rp);
    }
  }
  float* out = partial + (size_t)chunk * ts_count + j2;
  out[0] = acc0;
  if (pair) out[1] = acc1;
}

__global__ void k_time_series_combine(const float* __restrict__ partial,
                                      size_t ts_count, int n_chunks,
                                      float* __restrict__ ts) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x; j < ts_count;
       j += stride) {
    float acc = 0.0f;
    for (int c = 0; c < n_chunks; ++c) acc += partial[(size_t)c * ts_count + j];
    ts[j] = acc;
  }
}

// single-pass fallback (small shapes / no scratch)
__global__ void k_time_series(const float2* __restrict__ wf,
                              const uint8_t* __restrict__ flags, size_t rows,
                              size_t len, size_t ts_count,
                              float* __restrict__ ts) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x; j < ts_count;
       j += stride) {
    float acc = 0.0f;
    for (size_t i = 0; i < rows; ++i) {
      if (flags && flags[i]) continue;
      acc += norm2(wf[i * len + j]);
    }
    ts[j] = acc;
  }
}

__global__ void k_subtract_mean(float* __restrict__ ts, size_t n,
                                const double* __restrict__ sum) {
  const float mean = (float)(*sum / (double)n);
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    ts[i] -= mean;
}

__global__ void k_count_above(const float* __restrict__ ts, size_t n,
                              const double* __restrict__ sumsq, float snr,
                              unsigned* __restrict__ out_count,
                              float* __restrict__ out_threshold) {
  const float thr = snr * (float)sqrt(*sumsq / (double)n);
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  unsigned local = 0;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    local += (ts[i] > thr) ? 1u : 0u;
  // per-wave then per-block aggregation before one atomic (G12)
  __shared__ unsigned lds[kBlock / 64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off, 64);
  if (lane == 0) lds[wave] = local;
  __syncthreads();
  if (wave == 0) {
    unsigned v = (lane < kBlock / 64) ? lds[lane] : 0u;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
    if (lane == 0 && v) atomicAdd(out_count, v);
  }
  if (blockIdx.x == 0 && threadIdx.x == 0 && out_threshold) *out_threshold = thr;
}

// ---------------- inclusive scan (two-level) ----------------

constexpr int kScanItems = 8;  // items per thread → 2048 per block

__global__ void k_scan_blocks(const float* __restrict__ in,
                              float* __restrict__ out,
                              float* __restrict__ block_sums, size_t n) {
  __shared__ float lds[kBlock];
  const size_t base = (size_t)blockIdx.x * kBlock * kScanItems;
  float vals[kScanItems];
  float thread_sum = 0.0f;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    const size_t i = base + (size_t)threadIdx.x * kScanItems + k;
    const float v = (i < n) ? in[i] : 0.0f;
    thread_sum += v;
    vals[k] = thread_sum;  // inclusive within thread
  }
  // exclusive scan of per-thread sums across the block
  lds[threadIdx.x] = thread_sum;
  __syncthreads();
  // simple Hillis-Steele in LDS (block=256, 8 steps)
  float x = thread_sum;
#pragma unroll
  for (int off = 1; off < kBlock; off <<= 1) {
    float y = (threadIdx.x >= (unsigned)off) ? lds[threadIdx.x - off] : 0.0f;
    __syncthreads();
    x += y;
    lds[threadIdx.x] = x;
    __syncthreads();
  }
  const float excl = x - thread_sum;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    const size_t i = base + (size_t)threadIdx.x * kScanItems + k;
    if (i < n) out[i] = vals[k] + excl;
  }
  if (threadIdx.x == kBlock - 1 && block_sums) block_sums[blockIdx.x] = x;
}

__global__ void k_scan_block_sums(float* __restrict__ sums, int nb) {
  // single block: exclusive scan of up to 2048 block sums (serial per thread
  // chunk is fine: nb <= 2048 for n <= 4M; waterfall ts is ~2^18)
  __shared__ float lds[kBlock];
  float acc = 0.0f;
  const int per = (nb + kBlock - 1) / kBlock;
  const int lo = threadIdx.x * per;
  for (int i = lo; i < min(lo + per, nb); ++i) acc += sums[i];
  lds[threadIdx.x] = acc;
  __syncthreads();
  float x = acc;
  for (int off = 1; off < kBlock; off <<= 1) {
    float y = (threadIdx.x >= (unsigned)off) ? lds[threadIdx.x - off] : 0.0f;
    __syncthreads();
    x += y;
    lds[threadIdx.x] = x;
    __syncthreads();
  }
  const float excl_base = x - acc;
  float run = excl_base;
  for (int i = lo; i < min(lo + per, nb); ++i) {
    const float v = sums[i];
    sums[i] = run;  // exclusive
    run += v;
  }
}

__global__ void k_scan_add_offsets(float* __restrict__ out,
                                   const float* __restrict__ block_offsets,
                                   size_t n) {
  const size_t base = (size_t)blockIdx.x * kBlock * kScanItems;
  const float off = block_offsets[blockIdx.x];
  if (off == 0.0f) return;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    const size_t i = base + (size_t)threadIdx.x * kScanItems + k;
    if (i < n) out[i] += off;
  }
}

__global__ void k_boxcar(const float* __restrict__ cumsum,
                         float* __restrict__ out, size_t n_out, size_t L) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += stride)
    out[i] = cumsum[i + L] - cumsum[i];
}

// ---- fused boxcar ladder (3 launches instead of 3 per length) ----
// The per-length box series box_L[i] = cumsum[i+L] - cumsum[i] never
// materializes: stats, thresholds and counts all derive from the (L2-
// resident, ~1 MB) cumulative sum directly.

struct BoxLadderDev {
  int n;                        // number of ladder lengths (<= 12)
  unsigned long long L[12];
};

__global__ void k_box_stats(const float* __restrict__ cumsum, size_t ts_count,
                            BoxLadderDev lad, double* __restrict__ partials) {
  double s[12], s2[12];
#pragma unroll
  for (int l = 0; l < 12; ++l) s[l] = s2[l] = 0.0;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < ts_count; i += stride) {
    const float c0 = cumsum[i];
    for (int l = 0; l < lad.n; ++l) {
      const unsigned long long L = lad.L[l];
      if (i + L < ts_count) {
        const float v = cumsum[i + L] - c0;
        s[l] += v;
        s2[l] += (double)v * v;
      }
    }
  }
  for (int l = 0; l < lad.n; ++l) {
    const double bs = block_reduce_sum(s[l]);
    const double bs2 = block_reduce_sum(s2[l]);
    if (threadIdx.x == 0) {
      partials[(size_t)(2 * l) * kReducePartials + blockIdx.x] = bs;
      partials[(size_t)(2 * l + 1) * kReducePartials + blockIdx.x] = bs2;
    }
  }
}

__global__ void k_box_finish(const double* __restrict__ partials,
                             size_t ts_count, BoxLadderDev lad, float snr,
                             float* __restrict__ out_thr) {
  for (int l = 0; l < lad.n; ++l) {
    double s2 = 0.0;
    for (int i = threadIdx.x; i < kReducePartials; i += blockDim.x)
      s2 += partials[(size_t)(2 * l + 1) * kReducePartials + i];
    const double bs2 = block_reduce_sum(s2);
    if (threadIdx.x == 0) {
      const double n_out = (double)(ts_count - lad.L[l]);
      out_thr[l] = snr * (float)sqrt(bs2 / n_out);
    }
  }
}

__global__ void k_box_count(const float* __restrict__ cumsum, size_t ts_count,
                            BoxLadderDev lad,
                            const float* __restrict__ thr,
                            unsigned* __restrict__ out_counts) {
  float t[12];
  for (int l = 0; l < lad.n; ++l) t[l] = thr[l];
  unsigned cnt[12];
#pragma unroll
  for (int l = 0; l < 12; ++l) cnt[l] = 0;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < ts_count; i += stride) {
    const float c0 = cumsum[i];
    for (int l = 0; l < lad.n; ++l) {
      const unsigned long long L = lad.L[l];
      if (i + L < ts_count) cnt[l] += ((cumsum[i + L] - c0) > t[l]) ? 1u : 0u;
    }
  }
  __shared__ unsigned lds[kBlock / 64];
  for (int l = 0; l < lad.n; ++l) {
    unsigned local = cnt[l];
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      local += __shfl_down(local, off, 64);
    __syncthreads();
    if (lane == 0) lds[wave] = local;
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned b = 0;
      for (int w = 0; w < (int)(blockDim.x >> 6); ++w) b += lds[w];
      if (b) atomicAdd(&out_counts[l], b);
    }
  }
}

}  // namespace

// ---------------- host wrappers ----------------

int reduce_partials() { return 2 * kReducePartials; }

hipError_t mean_power(const float2* in, size_t n, double* partials,
                      double* out_mean, hipStream_t stream) {
  hipLaunchKernelGGL(k_mean_power_pass1, dim3(kReducePartials), dim3(kBlock),
                     0, stream, in, n, partials);
  hipLaunchKernelGGL(k_reduce_partials_mean, dim3(1), dim3(kBlock), 0, stream,
                     partials, kReducePartials, n, out_mean);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sum_sumsq(const float* in, size_t n, double* partials, double* out2,
                     hipStream_t stream) {
  hipLaunchKernelGGL(k_sum_sumsq_pass1, dim3(kReducePartials), dim3(kBlock), 0,
                     stream, in, n, partials);
  hipLaunchKernelGGL(k_reduce_partials_2, dim3(1), dim3(kBlock), 0, stream,
                     partials, kReducePartials, out2);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t rfi_s1(float2* spec, size_t n, const double* mean_p, float threshold,
                  float norm_coeff, hipStream_t stream) {
  hipLaunchKernelGGL(k_rfi_s1, grid_for(n), dim3(kBlock), 0, stream, spec, n,
                     mean_p, threshold, norm_coeff);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t zap_bins(float2* spec, size_t lo, size_t hi, hipStream_t stream) {
  if (hi < lo) return hipErrorInvalidValue;
  const size_t count = hi - lo + 1;
  hipLaunchKernelGGL(k_zap_bins, grid_for(count), dim3(kBlock), 0, stream,
                     spec, lo, count);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t dedisperse(float2* spec, size_t n, double f_min, double f_c,
                      double df, double dm, hipStream_t stream) {
  hipLaunchKernelGGL(k_dedisperse, grid_for(n), dim3(kBlock), 0, stream, spec,
                     n, f_min, f_c, df, dm);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t dedisp_phase_table(float2* table, size_t n, double f_min,
                              double f_c, double df, double dm,
                              hipStream_t stream) {
  hipLaunchKernelGGL(k_phase_table, grid_for(n), dim3(kBlock), 0, stream,
                     table, n, f_min, f_c, df, dm);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t rfi_dedisperse_fused(float2* spec, size_t n, const double* mean_p,
                                float threshold, float norm_coeff,
                                const ZapRange* ranges, int n_ranges,
                                double f_min, double f_c, double df, double dm,
                                const float2* factor_table,
                                hipStream_t stream) {
  if (n_ranges > 16) return hipErrorInvalidValue;
  ZapRangesArg za{};
  za.count = n_ranges;
  for (int i = 0; i < n_ranges; ++i) za.r[i] = ranges[i];
  const bool rfi = mean_p != nullptr;
  const dim3 g = grid_for(n);
  if (factor_table) {
    if (rfi)
      hipLaunchKernelGGL((k_rfi_dedisp_fused<true, true>), g, dim3(kBlock), 0,
                         stream, spec, n, mean_p, threshold, norm_coeff, za,
                         f_min, f_c, df, dm, factor_table);
    else
      hipLaunchKernelGGL((k_rfi_dedisp_fused<true, false>), g, dim3(kBlock), 0,
                         stream, spec, n, mean_p, threshold, norm_coeff, za,
                         f_min, f_c, df, dm, factor_table);
  } else {
    if (rfi)
      hipLaunchKernelGGL((k_rfi_dedisp_fused<false, true>), g, dim3(kBlock), 0,
                         stream, spec, n, mean_p, threshold, norm_coeff, za,
                         f_min, f_c, df, dm, factor_table);
    else
      hipLaunchKernelGGL((k_rfi_dedisp_fused<false, false>), g, dim3(kBlock),
                         0, stream, spec, n, mean_p, threshold, norm_coeff, za,
                         f_min, f_c, df, dm, factor_table);
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_row_stats(const float2* wf, size_t rows, size_t len,
                        float2* s2s4, hipStream_t stream) {
  if (len % 2 == 0)
    hipLaunchKernelGGL(k_sk_row_stats, dim3((uint32_t)rows), dim3(kBlock), 0,
                       stream, wf, len, s2s4);
  else
    hipLaunchKernelGGL(k_sk_row_stats_scalar, dim3((uint32_t)rows),
                       dim3(kBlock), 0, stream, wf, len, s2s4);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_flags(const float2* wf, const float2* s2s4, size_t rows,
                    size_t len, float lo_, float hi_, uint8_t* flags,
                    unsigned* zero_count, hipStream_t stream) {
  hipLaunchKernelGGL(k_sk_flags, grid_for(rows), dim3(kBlock), 0, stream, wf,
                     s2s4, rows, len, lo_, hi_, flags, zero_count);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

// K21: waterfall FFT-window de-apply (reference fft_pipe.hpp:350-358):
// wf[i] /= coef[i mod len] after the backward waterfall FFT, only when the
// configured FFT window is not the rectangle.  One elementwise pass.
__global__ void k_window_deapply(float2* __restrict__ wf,
                                 const float* __restrict__ coef, size_t total,
                                 size_t len) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const float inv = 1.0f / coef[i % len];
    const float2 x = wf[i];
    wf[i] = make_float2(x.x * inv, x.y * inv);
  }
}

hipError_t window_deapply(float2* wf, const float* coef, size_t total,
                          size_t len, hipStream_t stream) {
  hipLaunchKernelGGL(k_window_deapply, grid_for(total), dim3(kBlock), 0,
                     stream, wf, coef, total, len);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_zap_rows(float2* wf, const uint8_t* flags, size_t rows,
                       size_t len, hipStream_t stream) {
  if (rows > 65535) return hipErrorInvalidValue;
  dim3 grid(grid_for(len).x, (uint32_t)rows);
  hipLaunchKernelGGL(k_sk_zap_rows, grid, dim3(kBlock), 0, stream, wf, flags,
                     rows, len);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_v1_stats(const float2* wf, size_t M, size_t bins, float2* s2s4,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_sk_v1_stats, grid_for(bins), dim3(kBlock), 0, stream,
                     wf, M, bins, s2s4);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_v1_zap(float2* wf, size_t M, size_t bins, const float2* s2s4,
                     float lo_, float hi_, bool normalize,
                     hipStream_t stream) {
  if (normalize)
    hipLaunchKernelGGL((k_sk_v1_zap<true>), grid_for(M * bins), dim3(kBlock),
                       0, stream, wf, M, bins, s2s4, lo_, hi_);
  else
    hipLaunchKernelGGL((k_sk_v1_zap<false>), grid_for(M * bins), dim3(kBlock),
                       0, stream, wf, M, bins, s2s4, lo_, hi_);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t time_series(const float2* wf, const uint8_t* flags, size_t rows,
                       size_t len, size_t ts_count, float* ts,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_time_series, grid_for(ts_count), dim3(kBlock), 0,
                     stream, wf, flags, rows, len, ts_count, ts);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

int time_series_chunks(size_t ts_count) {
  const size_t grid_x = (ts_count / 2 + kBlock - 1) / kBlock;
  size_t c = 4096 / (grid_x ? grid_x : 1);
  if (c < 1) c = 1;
  if (c > 64) c = 64;
  return (int)c;
}

hipError_t time_series_2stage(const float2* wf, const uint8_t* flags,
                              size_t rows, size_t len, size_t ts_count,
                              float* ts, float* partial, hipStream_t stream) {
  if (len % 2 != 0 || ts_count < 2)
    return time_series(wf, flags, rows, len, ts_count, ts, stream);
  int chunks = time_series_chunks(ts_count);
  if ((size_t)chunks > rows) chunks = (int)rows;
  const uint32_t gx = (uint32_t)((ts_count / 2 + kBlock - 1) / kBlock);
  dim3 grid(gx, (uint32_t)chunks);
  if (flags)
    hipLaunchKernelGGL((k_time_series_partial<true>), grid, dim3(kBlock), 0,
                       stream, wf, flags, rows, len, ts_count, partial,
                       chunks);
  else
    hipLaunchKernelGGL((k_time_series_partial<false>), grid, dim3(kBlock), 0,
                       stream, wf, flags, rows, len, ts_count, partial,
                       chunks);
  hipLaunchKernelGGL(k_time_series_combine, grid_for(ts_count), dim3(kBlock),
                     0, stream, partial, ts_count, chunks, ts);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t subtract_mean(float* ts, size_t n, const double* sum,
                         hipStream_t stream) {
  hipLaunchKernelGGL(k_subtract_mean, grid_for(n), dim3(kBlock), 0, stream, ts,
                     n, sum);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t count_above(const float* ts, size_t n, const double* sumsq,
                       float snr, unsigned* out_count, float* out_threshold,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_count_above, grid_for(n), dim3(kBlock), 0, stream, ts,
                     n, sumsq, snr, out_count, out_threshold);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

int scan_scratch_size(size_t n) {
  const size_t per_block = (size_t)kBlock * kScanItems;
  return (int)((n + per_block - 1) / per_block);
}

hipError_t inclusive_scan(const float* in, float* out, size_t n,
                          float* scratch, hipStream_t stream) {
  const size_t per_block = (size_t)kBlock * kScanItems;
  const size_t nb = (n + per_block - 1) / per_block;
  // the block-sums pass is one workgroup scanning nb partials serially per
  // thread chunk — correct for any nb (scratch must hold nb floats)
  hipLaunchKernelGGL(k_scan_blocks, dim3((uint32_t)nb), dim3(kBlock), 0,
                     stream, in, out, scratch, n);
  hipLaunchKernelGGL(k_scan_block_sums, dim3(1), dim3(kBlock), 0, stream,
                     scratch, (int)nb);
  hipLaunchKernelGGL(k_scan_add_offsets, dim3((uint32_t)nb), dim3(kBlock), 0,
                     stream, out, scratch, n);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t boxcar(const float* cumsum, float* out, size_t n_out, size_t L,
                  hipStream_t stream) {
  hipLaunchKernelGGL(k_boxcar, grid_for(n_out), dim3(kBlock), 0, stream,
                     cumsum, out, n_out, L);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t boxcar_ladder(const float* cumsum, size_t ts_count,
                         const size_t* lengths, int n_lengths,
                         double* partials, float snr, float* out_thr,
                         unsigned* out_counts, hipStream_t stream) {
  if (n_lengths <= 0) return hipSuccess;
  if (n_lengths > 12) return hipErrorInvalidValue;
  BoxLadderDev lad{};
  lad.n = n_lengths;
  for (int l = 0; l < n_lengths; ++l)
    lad.L[l] = (unsigned long long)lengths[l];
  hipLaunchKernelGGL(k_box_stats, dim3(kReducePartials), dim3(kBlock), 0,
                     stream, cumsum, ts_count, lad, partials);
  hipLaunchKernelGGL(k_box_finish, dim3(1), dim3(kBlock), 0, stream,
                     partials, ts_count, lad, snr, out_thr);
  hipLaunchKernelGGL(k_box_count, grid_for(ts_count), dim3(kBlock), 0,
                     stream, cumsum, ts_count, lad, out_thr, out_counts);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

}  // namespace srtb_hip
