// Hand-written LDS-staged Stockham FFT for CDNA4 (gfx950).
//
// One generic pass kernel covers every shape the pipeline needs:
//  - batched contiguous C2C (any pow2 length ≤ kFftMaxLen per pass),
//  - four-step composite passes: strided column FFTs with fused inter-pass
//    twiddles (split-table exact), and row FFTs whose digit-reversal output
//    scatter replaces rocFFT's separate transpose kernels entirely,
//  - the r2c post-process (packed-real trick) as a separate small kernel.
//
// Index math is the NumPy oracle in srtb_amd/fftref.py (fft_small /
// fft_four_step / fft_deep / r2c_post); conventions are cuFFT's (forward
// sign -1, backward +1, unnormalized).
//
// Design notes (MI355X):
//  - Stockham auto-sort ping-pong in LDS: no bit-reversal permutation, so
//    global loads/stores stay linear per instance group and LDS access is
//    stride-regular; rows padded by one float2 to break power-of-2 bank
//    strides (G4).
//  - each workgroup (256 threads) processes F = ELEMS/n FFT instances; for
//    strided passes instances are q0-consecutive so a row of F elements is
//    contiguous in HBM (F*8 bytes per transaction run).
//  - butterfly twiddles come from a per-length global table (L2-resident,
//    broadcast reads); inter-pass twiddles e^{s*2πi*m/M} are exact via
//    split tables THi[m>>b]*TLo[m&(2^b-1)] computed in fp64 at plan build.

#include "common.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

namespace {

__device__ inline float2 cmulf(float2 a, float2 b) {
  return make_float2(a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x);
}

}  // namespace

// descriptor for one FFT pass (see srtb_kernels.h for the field contract)
struct FftPassDescDev {
  uint32_t n;            // pow2 FFT length of this pass
  uint32_t d0, d1;       // instance id -> q0 = id%d0, q1=(id/d0)%d1, q2=rest
  unsigned long long in_c0, in_c1, in_c2;    // input base = Σ q_i * c_i
  unsigned long long in_stride;
  unsigned long long out_c0, out_c1, out_c2;
  unsigned long long out_stride;
  unsigned long long tw_f0, tw_f1;  // twiddle factor = q0*tw_f0 + q1*tw_f1
  unsigned long long tw_mask;       // modulus-1 (modulus = pow2), 0 = off
  int tw_lo_bits;
};

namespace {

template <bool LOAD_FFAST, bool STORE_FFAST, bool TWIDDLE>
__global__ void __launch_bounds__(256)
    k_fft_stockham(const float2* __restrict__ in, float2* __restrict__ out,
                   FftPassDescDev d, int F,
                   const float2* __restrict__ tw_n,
                   const float2* __restrict__ tw_hi,
                   const float2* __restrict__ tw_lo) {
  extern __shared__ float2 lds[];
  const int n = d.n;
  const int ldst = n + 2;  // padded row stride (breaks pow2 bank conflicts)
  float2* X = lds;
  float2* Y = lds + (size_t)F * ldst;
  const unsigned long long fft0 = (unsigned long long)blockIdx.x * F;
  const int total = F * n;

  // ---- load ----
  for (int e = threadIdx.x; e < total; e += blockDim.x) {
    int f, i;
    if (LOAD_FFAST) { f = e % F; i = e / F; }
    else            { f = e / n; i = e % n; }
    const unsigned long long id = fft0 + f;
    const unsigned long long q0 = id % d.d0;
    const unsigned long long r = id / d.d0;
    const unsigned long long q1 = r % d.d1;
    const unsigned long long q2 = r / d.d1;
    const unsigned long long base =
        q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
    X[f * ldst + i] = in[base + (unsigned long long)i * d.in_stride];
  }
  __syncthreads();

  // ---- Stockham stages (radix-2 ping-pong; fftref.fft_small/fft0 math) ----
  const int half = n >> 1;
  int tstep = 1;  // n / ncur
  for (int ncur = n, s = 1; ncur > 1; ncur >>= 1, s <<= 1, tstep <<= 1) {
    const int m = ncur >> 1;
    for (int b = threadIdx.x; b < F * half; b += blockDim.x) {
      const int f = b / half;
      const int bb = b - f * half;
      const int p = bb / s;
      const int q = bb - p * s;
      const float2 a = X[f * ldst + q + s * p];
      const float2 c = X[f * ldst + q + s * (p + m)];
      const float2 w = tw_n[(size_t)p * tstep];
      const float2 diff = make_float2(a.x - c.x, a.y - c.y);
      Y[f * ldst + q + s * (2 * p)] = make_float2(a.x + c.x, a.y + c.y);
      Y[f * ldst + q + s * (2 * p + 1)] = cmulf(diff, w);
    }
    __syncthreads();
    float2* t = X;
    X = Y;
    Y = t;
  }

  // ---- store (+ inter-pass twiddle) ----
  for (int e = threadIdx.x; e < total; e += blockDim.x) {
    int f, k;
    if (STORE_FFAST) { f = e % F; k = e / F; }
    else             { f = e / n; k = e % n; }
    const unsigned long long id = fft0 + f;
    const unsigned long long q0 = id % d.d0;
    const unsigned long long r = id / d.d0;
    const unsigned long long q1 = r % d.d1;
    const unsigned long long q2 = r / d.d1;
    float2 v = X[f * ldst + k];
    if constexpr (TWIDDLE) {
      const unsigned long long tf = q0 * d.tw_f0 + q1 * d.tw_f1;
      const unsigned long long m_ = (tf * (unsigned long long)k) & d.tw_mask;
      const float2 w = cmulf(tw_hi[m_ >> d.tw_lo_bits],
                             tw_lo[m_ & ((1ull << d.tw_lo_bits) - 1)]);
      v = cmulf(v, w);
    }
    const unsigned long long base =
        q0 * d.out_c0 + q1 * d.out_c1 + q2 * d.out_c2;
    out[base + (unsigned long long)k * d.out_stride] = v;
  }
}

// twiddle-table builders (fp64 on device)
__global__ void k_build_twiddle(float2* __restrict__ t, size_t count,
                                double sign_two_pi_over_m) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x; j < count;
       j += stride) {
    double s, c;
    sincos(sign_two_pi_over_m * (double)j, &s, &c);
    t[j] = make_float2((float)c, (float)s);
  }
}

// r2c post-process (packed-real trick; fftref.r2c_post):
//   E = (Z[k]+conj(Z[M-k]))/2,  O = -i/2*(Z[k]-conj(Z[M-k])),
//   w(k) = exp(-2πi k/(2M)),
//   X[k] = E + w*O  and  X[M-k] = conj(E - w*O),
// so each thread handles the (k, M-k) PAIR — this makes the kernel
// in-place-safe (x may alias z).  k=0 gives X[0]=Re(Z0)+Im(Z0) (the would-be
// X[M] Nyquist bin is dropped, matching the pipeline count Nc).  Optionally
// accumulates Σ|X|² partials (fused mean-power for RFI s1 — saves a full
// 4 GB spectrum read).
template <bool MEANP>
__global__ void k_r2c_post(const float2* __restrict__ z,
                           float2* __restrict__ x, size_t m,
                           double* __restrict__ partials) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const size_t half = m / 2;
  double acc = 0.0;
  for (size_t k = (size_t)blockIdx.x * blockDim.x + threadIdx.x; k <= half;
       k += stride) {
    const float2 zk = z[k];
    const float2 zm = z[k == 0 ? 0 : m - k];
    const float2 zmc = make_float2(zm.x, -zm.y);
    const float2 even = make_float2(0.5f * (zk.x + zmc.x),
                                    0.5f * (zk.y + zmc.y));
    // odd = -0.5i * (zk - conj(z[m-k]))
    const float2 dif = make_float2(zk.x - zmc.x, zk.y - zmc.y);
    const float2 odd = make_float2(0.5f * dif.y, -0.5f * dif.x);
    double sw, cw;
    sincos(-M_PI * (double)k / (double)m, &sw, &cw);
    const float2 w = make_float2((float)cw, (float)sw);
    const float2 wo = cmulf(w, odd);
    const float2 xk = make_float2(even.x + wo.x, even.y + wo.y);
    x[k] = xk;
    if constexpr (MEANP) acc += (double)xk.x * xk.x + (double)xk.y * xk.y;
    if (k != 0 && k != half) {
      const float2 xm = make_float2(even.x - wo.x, -(even.y - wo.y));
      x[m - k] = xm;
      if constexpr (MEANP)
        acc += (double)xm.x * xm.x + (double)xm.y * xm.y;
    }
  }
  if constexpr (MEANP) {
    const double b = block_reduce_sum(acc);
    if (threadIdx.x == 0) partials[blockIdx.x] = b;
  }
}

__global__ void k_r2c_post_finish_mean(const double* __restrict__ partials,
                                       int np, size_t n,
                                       double* __restrict__ out_mean) {
  double acc = 0.0;
  for (int i = threadIdx.x; i < np; i += blockDim.x) acc += partials[i];
  const double b = block_reduce_sum(acc);
  if (threadIdx.x == 0) *out_mean = b / (double)n;
}

constexpr int kR2cPostBlocks = 1024;

}  // namespace

hipError_t fft_build_twiddle(float2* table, size_t count, double m, int sign,
                             hipStream_t stream) {
  hipLaunchKernelGGL(k_build_twiddle, grid_for(count), dim3(kBlock), 0,
                     stream, table, count, sign * 2.0 * M_PI / m);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t fft_stockham_pass(const float2* in, float2* out,
                             const FftPassDesc& hd, size_t n_ffts, int F,
                             bool load_ffast, bool store_ffast,
                             const float2* tw_n, const float2* tw_hi,
                             const float2* tw_lo, hipStream_t stream) {
  FftPassDescDev d;
  d.n = hd.n;
  d.d0 = hd.d0;
  d.d1 = hd.d1;
  d.in_c0 = hd.in_c0; d.in_c1 = hd.in_c1; d.in_c2 = hd.in_c2;
  d.in_stride = hd.in_stride;
  d.out_c0 = hd.out_c0; d.out_c1 = hd.out_c1; d.out_c2 = hd.out_c2;
  d.out_stride = hd.out_stride;
  d.tw_f0 = hd.tw_f0; d.tw_f1 = hd.tw_f1;
  d.tw_mask = hd.tw_mod ? hd.tw_mod - 1 : 0;
  d.tw_lo_bits = hd.tw_lo_bits;
  const bool twiddle = hd.tw_mod != 0;
  if (n_ffts % F != 0) return hipErrorInvalidValue;
  const uint32_t grid = (uint32_t)(n_ffts / F);
  const size_t lds_bytes = 2ull * F * (hd.n + 2) * sizeof(float2);
  if (lds_bytes > 160 * 1024) return hipErrorInvalidValue;

#define DISPATCH(LF, SF, TW)                                                \
  hipLaunchKernelGGL((k_fft_stockham<LF, SF, TW>), dim3(grid), dim3(256),   \
                     lds_bytes, stream, in, out, d, F, tw_n, tw_hi, tw_lo)
  if (load_ffast) {
    if (store_ffast) { if (twiddle) DISPATCH(true, true, true); else DISPATCH(true, true, false); }
    else             { if (twiddle) DISPATCH(true, false, true); else DISPATCH(true, false, false); }
  } else {
    if (store_ffast) { if (twiddle) DISPATCH(false, true, true); else DISPATCH(false, true, false); }
    else             { if (twiddle) DISPATCH(false, false, true); else DISPATCH(false, false, false); }
  }
#undef DISPATCH
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t r2c_post_process(const float2* z, float2* x, size_t m,
                            double* mean_partials, double* out_mean,
                            hipStream_t stream) {
  if (mean_partials) {
    hipLaunchKernelGGL((k_r2c_post<true>), dim3(kR2cPostBlocks), dim3(kBlock),
                       0, stream, z, x, m, mean_partials);
    hipLaunchKernelGGL(k_r2c_post_finish_mean, dim3(1), dim3(kBlock), 0,
                       stream, mean_partials, kR2cPostBlocks, m, out_mean);
  } else {
    hipLaunchKernelGGL((k_r2c_post<false>), dim3(kR2cPostBlocks), dim3(kBlock),
                       0, stream, z, x, m, nullptr);
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

}  // namespace srtb_hip
