// Hand-written LDS-staged Stockham FFT for CDNA4 (gfx950).
//
// One generic pass kernel covers every shape the pipeline needs:
//  - batched contiguous C2C (any pow2 length ≤ 4096 per pass),
//  - four-step composite passes: strided column FFTs with fused inter-pass
//    twiddles (split-table exact), and row FFTs whose digit-reversal output
//    scatter replaces rocFFT's separate transpose kernels entirely,
//  - the r2c post-process (packed-real trick) as a separate small kernel.
//
// Index math is the NumPy oracle in srtb_amd/fftref.py (fft_small_r4 /
// fft_four_step / fft_deep / r2c_post); conventions are cuFFT's (forward
// sign -1, backward +1, unnormalized).
//
// Design notes (MI355X):
//  - Stockham auto-sort ping-pong in LDS: no bit-reversal permutation, so
//    global loads/stores stay linear per instance group and LDS access is
//    stride-regular; rows padded by one float2 to break pow2 bank strides.
//  - mixed radix-4 stages (one final radix-2 for odd log2) halve LDS traffic
//    and barrier count vs radix-2.
//  - ALL addressing is shift/mask (every radix, digit and group count is a
//    power of two): per-element 64-bit divisions cost ~10x on these loops.
//  - the per-length butterfly twiddle table (full circle, n entries) is
//    staged into LDS once per workgroup; inter-pass twiddles
//    e^{s*2πi*m/M} are exact via split tables THi[m>>b]*TLo[m&(2^b-1)]
//    built in fp64 at plan time (L2-resident).
//  - each workgroup (256 threads) processes F = ELEMS/n FFT instances; for
//    strided passes instances are q0-consecutive so a row of F elements is
//    contiguous in HBM.

#include <cstdlib>
#include "common.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

namespace {

__device__ inline float2 cmulf(float2 a, float2 b) {
  return make_float2(a.x * b.x - a.y * b.y, a.x * b.y + a.y * b.x);
}

}  // namespace

// device-side pass descriptor; counts as log2 (see srtb_kernels.h contract)
struct FftPassDescDev {
  uint32_t n;
  int n_log2, f_log2;
  int d0_log2, d1_log2;
  unsigned long long in_c0, in_c1, in_c2;
  unsigned long long in_stride;
  unsigned long long out_c0, out_c1, out_c2;
  unsigned long long out_stride;
  unsigned long long tw_f0, tw_f1;
  unsigned long long tw_mask;  // modulus-1; twiddle enabled via template
  int tw_lo_bits;
  double tw_angle;  // sign * 2*pi / modulus
  // tuning experiments (host env switches): bit0 = XCD-aware workgroup
  // swizzle (8 XCDs dispatch round-robin by blockIdx; remapping gives each
  // XCD a contiguous column range), bit1 = non-temporal loads/stores
  uint32_t tuning = 0;
};

// XCD-aware workgroup remap: blockIdx b runs on XCD b%8, so b' =
// (b&7)*(G/8) + b>>3 hands XCD k the contiguous index range [k*G/8, ...).
__device__ inline unsigned int xcd_swizzle(unsigned int bx, unsigned int g,
                                           uint32_t tuning) {
  if ((tuning & 1u) && (g & 7u) == 0u)
    return (bx & 7u) * (g >> 3) + (bx >> 3);
  return bx;
}

// non-temporal (streaming) access experiment: every element of a column
// pass is touched exactly once, so L2 retention buys nothing within a pass
typedef float fvec2_t __attribute__((ext_vector_type(2)));
__device__ inline float2 ld_stream(const float2* p, uint32_t tuning) {
  if (tuning & 2u) {
    const fvec2_t v =
        __builtin_nontemporal_load(reinterpret_cast<const fvec2_t*>(p));
    return make_float2(v.x, v.y);
  }
  return *p;
}
__device__ inline void st_stream(float2* p, float2 v, uint32_t tuning) {
  if (tuning & 2u) {
    fvec2_t w;
    w.x = v.x;
    w.y = v.y;
    __builtin_nontemporal_store(w, reinterpret_cast<fvec2_t*>(p));
  } else {
    *p = v;
  }
}

namespace {

// inter-pass twiddle e^{i * m * tw_angle}: exact integer reduction (m < M),
// fp64 angle, fast f32 sincos — table gathers spread over up to 64 cache
// lines per wave instruction and were the column kernel's bottleneck.
__device__ inline float2 tw_eval(unsigned long long m, double angle) {
  const float th = (float)((double)m * angle);
  float s, c;
  __sincosf(th, &s, &c);
  return make_float2(c, s);
}

__device__ inline void digits(unsigned long long id, const FftPassDescDev& d,
                              unsigned long long& q0, unsigned long long& q1,
                              unsigned long long& q2) {
  q0 = id & ((1ull << d.d0_log2) - 1);
  const unsigned long long r = id >> d.d0_log2;
  q1 = r & ((1ull << d.d1_log2) - 1);
  q2 = r >> d.d1_log2;
}

// fused sub-byte unpack for the FORWARD first pass: complex element `flat`
// of the packed-real view is real samples (2*flat, 2*flat+1), which live in
// the same raw byte (MSB-first fields).  Reading 0.12-0.5 GB of bytes
// replaces a 4 GB float re-read AND the standalone unpack kernel's 4 GB
// write (measured +7.6%% end-to-end on the 2-bit J1644 config).
template <int NBITS>
__device__ inline float2 decN_load(const uint8_t* __restrict__ raw,
                                   unsigned long long flat) {
  if constexpr (NBITS == 4) {
    const uint32_t bv = raw[flat];
    return make_float2((float)((bv >> 4) & 15u), (float)(bv & 15u));
  } else if constexpr (NBITS == 2) {
    const uint32_t bv = raw[flat >> 1];
    const int j0 = (int)(flat & 1) * 2;
    return make_float2((float)((bv >> ((3 - j0) * 2)) & 3u),
                       (float)((bv >> ((2 - j0) * 2)) & 3u));
  } else if constexpr (NBITS == 8) {
    return make_float2((float)raw[2 * flat], (float)raw[2 * flat + 1]);
  } else if constexpr (NBITS == -8) {
    const int8_t* r8 = reinterpret_cast<const int8_t*>(raw);
    return make_float2((float)r8[2 * flat], (float)r8[2 * flat + 1]);
  } else if constexpr (NBITS == 16) {
    const uint16_t* r16 = reinterpret_cast<const uint16_t*>(raw);
    return make_float2((float)r16[2 * flat], (float)r16[2 * flat + 1]);
  } else if constexpr (NBITS == -16) {
    const int16_t* r16 = reinterpret_cast<const int16_t*>(raw);
    return make_float2((float)r16[2 * flat], (float)r16[2 * flat + 1]);
  } else {  // NBITS == 1
    const uint32_t bv = raw[flat >> 2];
    const int j0 = (int)(flat & 3) * 2;
    return make_float2((float)((bv >> (7 - j0)) & 1u),
                       (float)((bv >> (6 - j0)) & 1u));
  }
}

constexpr int col_ilog2(int v) { return v <= 1 ? 0 : 1 + col_ilog2(v / 2); }

// schedule: first-to-last radices = [4]*(t/2) + [2 if t odd];
// σ_N(n) = (n % r_last)*M + σ_M(n / r_last)
template <int N>
constexpr int col_sigma(int n) {
  int res = 0, x = n, cur = N;
  // iterate from the LAST stage radix down; last radix is 2 iff t odd
  int t = col_ilog2(N);
  bool odd = (t & 1) != 0;
  // stage radices last-to-first: if odd, radices are [2, 4, 4, ...] from last?
  // Schedule first-to-last: [4,4,...,2] (trailing 2).  Last radix = 2 if odd.
  bool first_iter = true;
  while (cur > 1) {
    int r;
    if (first_iter && odd) r = 2;
    else r = 4;
    first_iter = false;
    const int M = cur / r;
    res += (x % r) * M;
    x /= r;
    cur = M;
  }
  return res;
}

template <int SIGN>
__device__ inline void bfly2(float2& a, float2& b, float2 w) {
  // DIT radix-2: y1 = b*w; out: a+y1, a-y1
  const float2 y = cmulf(b, w);
  b = make_float2(a.x - y.x, a.y - y.y);
  a = make_float2(a.x + y.x, a.y + y.y);
}

template <int SIGN>
__device__ inline void bfly4(float2& a, float2& b, float2& c, float2& d,
                             float2 w1, float2 w2, float2 w3) {
  const float2 y1 = cmulf(b, w1);
  const float2 y2 = cmulf(c, w2);
  const float2 y3 = cmulf(d, w3);
  const float2 t0 = make_float2(a.x + y2.x, a.y + y2.y);
  const float2 t1 = make_float2(a.x - y2.x, a.y - y2.y);
  const float2 t2 = make_float2(y1.x + y3.x, y1.y + y3.y);
  const float2 dmy = make_float2(y1.x - y3.x, y1.y - y3.y);
  const float2 t3 = (SIGN > 0) ? make_float2(-dmy.y, dmy.x)
                               : make_float2(dmy.y, -dmy.x);
  a = make_float2(t0.x + t2.x, t0.y + t2.y);
  b = make_float2(t1.x + t3.x, t1.y + t3.y);
  c = make_float2(t0.x - t2.x, t0.y - t2.y);
  d = make_float2(t1.x - t3.x, t1.y - t3.y);
}

// one DIT stage over the register array; L = output block size, radix r.
// TS overrides the twiddle index scale when the register array holds only a
// fraction of the full column (the lane-pair kernel below).
template <int N, int L, int R, int SIGN, int TS = N / L>
__device__ inline void col_stage(float2 (&v)[N],
                                 const float2* __restrict__ tw_n) {
  constexpr int M = L / R;  // TS: W_L^x = tw_n[x*TS]
#pragma unroll
  for (int g = 0; g < N; g += L) {
#pragma unroll
    for (int j = 0; j < M; ++j) {
      if constexpr (R == 2) {
        const float2 w = (j == 0) ? make_float2(1.f, 0.f) : tw_n[j * TS];
        bfly2<SIGN>(v[g + j], v[g + j + M], w);
      } else {
        float2 w1, w2, w3;
        if (j == 0) {
          w1 = w2 = w3 = make_float2(1.f, 0.f);
        } else {
          w1 = tw_n[j * TS];
          w2 = tw_n[2 * j * TS];
          w3 = tw_n[3 * j * TS];
        }
        bfly4<SIGN>(v[g + j], v[g + j + M], v[g + j + 2 * M],
                    v[g + j + 3 * M], w1, w2, w3);
      }
    }
  }
}

template <int N, int SIGN>
__device__ inline void col_fft(float2 (&v)[N],
                               const float2* __restrict__ tw_n) {
  constexpr int T = col_ilog2(N);
  // first-to-last: radix-4 stages, trailing radix-2 when T is odd.
  if constexpr (T >= 2) col_stage<N, 4, 4, SIGN>(v, tw_n);
  if constexpr (T >= 4) col_stage<N, 16, 4, SIGN>(v, tw_n);
  if constexpr (T >= 6) col_stage<N, 64, 4, SIGN>(v, tw_n);
  if constexpr (T == 1) col_stage<N, 2, 2, SIGN>(v, tw_n);
  if constexpr (T == 3) col_stage<N, 8, 2, SIGN>(v, tw_n);
  if constexpr (T == 5) col_stage<N, 32, 2, SIGN>(v, tw_n);
}


// LDS layout: [tw: n][X: F*(n+2)][Y: F*(n+2)] float2s.
template <bool LOAD_FFAST, bool STORE_FFAST, bool TWIDDLE, int SIGN>
__global__ void __launch_bounds__(256)
    k_fft_stockham(const float2* __restrict__ in, float2* __restrict__ out,
                   FftPassDescDev d, const float2* __restrict__ tw_n,
                   const float2* __restrict__ tw_hi,
                   const float2* __restrict__ tw_lo) {
  extern __shared__ float2 lds[];
  const int n = d.n;
  const int nl = d.n_log2;
  const int F = 1 << d.f_log2;
  const int ldst = n + 2;
  float2* ltw = lds;
  float2* X = lds + n;
  float2* Y = X + (size_t)F * ldst;
  float2* ltw16 = Y + (size_t)F * ldst;  // 16-entry W_16 table (radix-16)
  const unsigned long long fft0 = (unsigned long long)blockIdx.x << d.f_log2;
  const int total = F << nl;

  for (int j = threadIdx.x; j < n; j += blockDim.x) ltw[j] = tw_n[j];
  if (nl >= 4 && threadIdx.x < 16)
    ltw16[threadIdx.x] = tw_n[(size_t)threadIdx.x << (nl - 4)];

  // ---- load (register-staged in chunks of 8 so ≥8 global loads stay in
  // flight per thread; a naive load→ds_write loop serializes on vmcnt) ----
  {
    const int iters = total >> 8;  // blockDim == 256
    int it = 0;
    for (; it + 8 <= iters; it += 8) {
      float2 tmp[8];
      int lidx[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const int e = ((it + k) << 8) + threadIdx.x;
        int f, i;
        if (LOAD_FFAST) { f = e & (F - 1); i = e >> d.f_log2; }
        else            { f = e >> nl; i = e & (n - 1); }
        unsigned long long q0, q1, q2;
        digits(fft0 + f, d, q0, q1, q2);
        const unsigned long long base =
            q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
        tmp[k] = in[base + (unsigned long long)i * d.in_stride];
        lidx[k] = f * ldst + i;
      }
#pragma unroll
      for (int k = 0; k < 8; ++k) X[lidx[k]] = tmp[k];
    }
    for (int e = (it << 8) + threadIdx.x; e < total; e += blockDim.x) {
      int f, i;
      if (LOAD_FFAST) { f = e & (F - 1); i = e >> d.f_log2; }
      else            { f = e >> nl; i = e & (n - 1); }
      unsigned long long q0, q1, q2;
      digits(fft0 + f, d, q0, q1, q2);
      const unsigned long long base =
          q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
      X[f * ldst + i] = in[base + (unsigned long long)i * d.in_stride];
    }
  }
  __syncthreads();

  int ncur = n, s = 1, tstep_log2 = 0;
  // ---- radix-16 stages: one 16-point register FFT per butterfly (the
  // col_fft<16> machinery) — halves the LDS round trips + barriers vs
  // pure radix-4 (1024: 5 stages -> 16,16,4 = 3) ----
  if (d.tuning & 4u) {
    const int sixteenth = n >> 4;
    while ((ncur & 15) == 0 && ncur >= 16) {
      const int m = ncur >> 4;
      const int s_log2 = tstep_log2;
      for (int b = threadIdx.x; b < F * sixteenth; b += blockDim.x) {
        const int f = b >> (nl - 4);
        const int bb = b & (sixteenth - 1);
        const int p = bb >> s_log2;
        const int q = bb & (s - 1);
        const float2* row = X + f * ldst;
        float2 v[16];
#pragma unroll
        for (int j = 0; j < 16; ++j)
          v[col_sigma<16>(j)] = row[q + s * (p + j * m)];
        col_fft<16, SIGN>(v, ltw16);
        float2* orow = Y + f * ldst;
        const int i1 = p << tstep_log2;
        const int ob = q + ((p * s) << 4);
        orow[ob] = v[0];
#pragma unroll
        for (int j = 1; j < 16; ++j)
          orow[ob + j * s] = cmulf(v[j], ltw[i1 * j]);
      }
      __syncthreads();
      float2* tswp = X;
      X = Y;
      Y = tswp;
      ncur >>= 4;
      s <<= 4;
      tstep_log2 += 4;
    }
  }
  // ---- radix-4 stages (fftref.fft_small_r4 math) ----
  const int quarter = n >> 2;
  while ((ncur & 3) == 0 && ncur > 1) {
    const int m = ncur >> 2;
    const int s_log2 = tstep_log2;  // s == 1 << tstep_log2 here
    for (int b = threadIdx.x; b < F * quarter; b += blockDim.x) {
      const int f = b >> (nl - 2);
      const int bb = b & (quarter - 1);
      const int p = bb >> s_log2;
      const int q = bb & (s - 1);
      const float2* row = X + f * ldst;
      const float2 a = row[q + s * p];
      const float2 bv = row[q + s * (p + m)];
      const float2 c = row[q + s * (p + 2 * m)];
      const float2 dv = row[q + s * (p + 3 * m)];
      const float2 apc = make_float2(a.x + c.x, a.y + c.y);
      const float2 amc = make_float2(a.x - c.x, a.y - c.y);
      const float2 bpd = make_float2(bv.x + dv.x, bv.y + dv.y);
      const float2 bmd = make_float2(bv.x - dv.x, bv.y - dv.y);
      // si*(b-d) with si = SIGN*i: i*(x,y) = (-y, x)
      const float2 sibmd = (SIGN > 0) ? make_float2(-bmd.y, bmd.x)
                                      : make_float2(bmd.y, -bmd.x);
      const float2 u0 = make_float2(apc.x + bpd.x, apc.y + bpd.y);
      const float2 u2 = make_float2(apc.x - bpd.x, apc.y - bpd.y);
      const float2 u1 = make_float2(amc.x + sibmd.x, amc.y + sibmd.y);
      const float2 u3 = make_float2(amc.x - sibmd.x, amc.y - sibmd.y);
      const int i1 = p << tstep_log2;
      const float2 w1 = ltw[i1];
      const float2 w2 = ltw[i1 * 2];
      const float2 w3 = ltw[i1 * 3];
      float2* orow = Y + f * ldst;
      const int ob = q + ((p * s) << 2);
      orow[ob] = u0;
      orow[ob + s] = cmulf(u1, w1);
      orow[ob + 2 * s] = cmulf(u2, w2);
      orow[ob + 3 * s] = cmulf(u3, w3);
    }
    __syncthreads();
    float2* t = X; X = Y; Y = t;
    ncur >>= 2;
    s <<= 2;
    tstep_log2 += 2;
  }
  // ---- final radix-2 stage (odd log2(n)) ----
  if (ncur == 2) {
    const int half = n >> 1;  // == s at this point
    for (int b = threadIdx.x; b < F * half; b += blockDim.x) {
      const int f = b >> (nl - 1);
      const int q = b & (half - 1);
      const float2* row = X + f * ldst;
      const float2 a = row[q];
      const float2 c = row[q + half];
      Y[f * ldst + q] = make_float2(a.x + c.x, a.y + c.y);
      Y[f * ldst + q + half] = make_float2(a.x - c.x, a.y - c.y);
    }
    __syncthreads();
    float2* t = X; X = Y; Y = t;
  }

  // ---- store (+ inter-pass twiddle) ----
  for (int e = threadIdx.x; e < total; e += blockDim.x) {
    int f, k;
    if (STORE_FFAST) { f = e & (F - 1); k = e >> d.f_log2; }
    else             { f = e >> nl; k = e & (n - 1); }
    unsigned long long q0, q1, q2;
    digits(fft0 + f, d, q0, q1, q2);
    float2 v = X[f * ldst + k];
    if constexpr (TWIDDLE) {
      const unsigned long long tf = q0 * d.tw_f0 + q1 * d.tw_f1;
      const unsigned long long m_ = (tf * (unsigned long long)k) & d.tw_mask;
      v = cmulf(v, tw_eval(m_, d.tw_angle));
    }
    const unsigned long long base =
        q0 * d.out_c0 + q1 * d.out_c1 + q2 * d.out_c2;
    out[base + (unsigned long long)k * d.out_stride] = v;
  }
}

// ---------------------------------------------------------------------------
// Register-resident column FFT (one FFT per thread, N ≤ 64 in VGPRs).
//
// Used for the strided passes of large composites: a wave's 64 lanes own 64
// CONSECUTIVE columns, so every load/store instruction touches a contiguous
// 512-byte run regardless of the column stride — the LDS kernel's F-limited
// runs (64 B at F=8) were the bandwidth killer on these passes.
//
// Algorithm: mixed-radix in-place DIT (radix-4 stages + trailing radix-2),
// input permutation σ folded into the (compile-time) register indices —
// validated in srtb_amd/fftref.py and the dit_mixed prototype.
// ---------------------------------------------------------------------------

struct FftPreopDev {
  const double* mean_power;
  float threshold, norm_coeff;
  int n_zap;
  ZapRange zap[16];
  double f_min, f_c, df, dm;
  const float2* table;  // cached dedispersion factors (null = compute fp64)
  // != 0: the pass input is the PACKED forward spectrum Z (length r2c_m);
  // the r2c pair-combine X[k] = E + w_k O runs at load (k_r2c_post math),
  // eliminating the standalone r2c pass.  Requires an out-of-place first
  // pass: element k also reads Z[r2c_m - k].
  unsigned long long r2c_m;
};

// r2c pair-combine at load (identical math/precision to k_r2c_post)
__device__ inline float2 r2c_combine_load(const float2* __restrict__ zbase,
                                          float2 zk, unsigned long long k,
                                          unsigned long long m) {
  if (k == 0) return make_float2(zk.x + zk.y, 0.0f);
  const float2 zm = zbase[m - k];
  const float2 zmc = make_float2(zm.x, -zm.y);
  const float2 even = make_float2(0.5f * (zk.x + zmc.x),
                                  0.5f * (zk.y + zmc.y));
  const float2 dif = make_float2(zk.x - zmc.x, zk.y - zmc.y);
  const float2 odd = make_float2(0.5f * dif.y, -0.5f * dif.x);
  const float phiw = (float)(-M_PI * (double)k / (double)m);
  float sw, cw;
  __sincosf(phiw, &sw, &cw);
  const float2 wo = make_float2(cw * odd.x - sw * odd.y,
                                cw * odd.y + sw * odd.x);
  return make_float2(even.x + wo.x, even.y + wo.y);
}

template <int N, bool TWIDDLE, int SIGN, bool PREOP, int DEC = 0>
__global__ void __launch_bounds__(256)
    k_fft_col(const float2* __restrict__ in, float2* __restrict__ out,
              FftPassDescDev d, unsigned long long n_ffts,
              const float2* __restrict__ tw_n,
              const float2* __restrict__ tw_hi,
              const float2* __restrict__ tw_lo, FftPreopDev pre,
              const uint8_t* __restrict__ raw2) {
  const unsigned long long id =
      (unsigned long long)xcd_swizzle(blockIdx.x, gridDim.x, d.tuning) *
          blockDim.x + threadIdx.x;
  if (id >= n_ffts) return;
  unsigned long long q0, q1, q2;
  digits(id, d, q0, q1, q2);
  const unsigned long long base = q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
  float thr_mean = 0.f;
  if constexpr (PREOP) {
    if (pre.mean_power) thr_mean = pre.threshold * (float)(*pre.mean_power);
  }
  float2 v[N];
  // a column spans (N-1)*stride < L <= 2^29 elements, so 32-bit offsets
  // from the column base suffice — halves the in-flight address registers
  const float2* __restrict__ colbase = in + base;
  const uint32_t stride32 = (uint32_t)d.in_stride;
#pragma unroll
  for (int i = 0; i < N; ++i) {
    const uint32_t off = (uint32_t)i * stride32;
    const unsigned long long flat = base + off;
    float2 x;
    if constexpr (DEC != 0) x = decN_load<DEC>(raw2, flat);
    else x = ld_stream(colbase + off, d.tuning);
    if constexpr (PREOP) {
      // fused rfi_dedisperse (spectrum.hip k_rfi_dedisp_fused semantics);
      // `flat` IS the spectrum bin index for in-place column passes
      if (pre.r2c_m) x = r2c_combine_load(in, x, flat, pre.r2c_m);
      bool zap = pre.mean_power && (norm2(x) > thr_mean);
      for (int z = 0; z < pre.n_zap; ++z)
        zap |= (flat >= pre.zap[z].lo) & (flat <= pre.zap[z].hi);
      if (zap) {
        x = make_float2(0.f, 0.f);
      } else {
        if (pre.mean_power) {
          x.x *= pre.norm_coeff;
          x.y *= pre.norm_coeff;
        }
        x = cmulf(x, pre.table ? pre.table[flat]
                               : srtb_dedisp_factor(flat, pre.f_min, pre.f_c,
                                                    pre.df, pre.dm));
      }
    }
    v[col_sigma<N>(i)] = x;
  }
  col_fft<N, SIGN>(v, tw_n);
  float2* __restrict__ ocolbase = out + base;
#pragma unroll
  for (int k = 0; k < N; ++k) {
    float2 r = v[k];
    if constexpr (TWIDDLE) {
      const unsigned long long m_ =
          (q0 * d.tw_f0 * (unsigned long long)k) & d.tw_mask;
      r = cmulf(r, tw_eval(m_, d.tw_angle));
    }
    st_stream(ocolbase + (uint32_t)k * stride32, r, d.tuning);
  }
}

// ---------------------------------------------------------------------------
// N = 64 column pass, two lanes per column.  The mono-column k_fft_col<64>
// holds 64 complex elements in registers (~400 VGPRs → 1 wave/SIMD); here
// lane p = tid&1 of an adjacent lane pair holds post-σ positions
// 32p..32p+31, so stages 1-2 (L=4, L=16) are lane-local, and the final
// L=64 stage exchanges halves with two shfl_xor(…, 1) per butterfly (a
// quad-perm DPP swap on gfx950 — no LDS).  Both lanes evaluate the full
// radix-4 butterfly from the exchanged quartet and keep their own two
// outputs.  32 data registers per lane → ~3 waves/SIMD, and per-instruction
// global accesses stay coalesced (each half-wave touches 32 consecutive
// columns: 2 × 256 B runs).
// ---------------------------------------------------------------------------

constexpr int col_sigma_inv64(int pos) {
  for (int i = 0; i < 64; ++i)
    if (col_sigma<64>(i) == pos) return i;
  return 0;
}

__device__ inline float2 shfl_xor1(float2 x) {
  return make_float2(__shfl_xor(x.x, 1, 64), __shfl_xor(x.y, 1, 64));
}

template <bool TWIDDLE, int SIGN, bool PREOP, int DEC = 0>
__global__ void __launch_bounds__(256)
    k_fft_col_pair64(const float2* __restrict__ in, float2* __restrict__ out,
                     FftPassDescDev d, unsigned long long n_ffts,
                     const float2* __restrict__ tw_n,
                     const float2* __restrict__ tw_hi,
                     const float2* __restrict__ tw_lo, FftPreopDev pre,
                     const uint8_t* __restrict__ raw2) {
  const unsigned long long tid =
      (unsigned long long)xcd_swizzle(blockIdx.x, gridDim.x, d.tuning) *
          blockDim.x + threadIdx.x;
  const unsigned long long id = tid >> 1;  // column index
  const int p = (int)(tid & 1);            // which half of the column
  if (id >= n_ffts) return;
  unsigned long long q0, q1, q2;
  digits(id, d, q0, q1, q2);
  const unsigned long long base = q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
  float thr_mean = 0.f;
  if constexpr (PREOP) {
    if (pre.mean_power) thr_mean = pre.threshold * (float)(*pre.mean_power);
  }
  float2 v[32];
  const float2* __restrict__ colbase = in + base;
  const uint32_t stride32 = (uint32_t)d.in_stride;
#pragma unroll
  for (int r = 0; r < 32; ++r) {
    // position 32p + r holds element σ⁻¹(32p + r); both candidates are
    // compile-time constants, selected per-lane
    const int i = p ? col_sigma_inv64(32 + r) : col_sigma_inv64(r);
    const uint32_t off = (uint32_t)i * stride32;
    float2 x;
    if constexpr (DEC != 0) x = decN_load<DEC>(raw2, base + off);
    else x = ld_stream(colbase + off, d.tuning);
    if constexpr (PREOP) {
      const unsigned long long flat = base + off;
      if (pre.r2c_m) x = r2c_combine_load(in, x, flat, pre.r2c_m);
      bool zap = pre.mean_power && (norm2(x) > thr_mean);
      for (int z = 0; z < pre.n_zap; ++z)
        zap |= (flat >= pre.zap[z].lo) & (flat <= pre.zap[z].hi);
      if (zap) {
        x = make_float2(0.f, 0.f);
      } else {
        if (pre.mean_power) {
          x.x *= pre.norm_coeff;
          x.y *= pre.norm_coeff;
        }
        x = cmulf(x, pre.table ? pre.table[flat]
                               : srtb_dedisp_factor(flat, pre.f_min, pre.f_c,
                                                    pre.df, pre.dm));
      }
    }
    v[r] = x;
  }
  // stages 1-2 are lane-local; the twiddle scale is that of the FULL
  // length-64 transform (TS = 64/L), not of the 32-element register array
  col_stage<32, 4, 4, SIGN, 16>(v, tw_n);
  col_stage<32, 16, 4, SIGN, 4>(v, tw_n);
  // stage 3: quartets {j, j+16, j+32, j+48}, TS = 1
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const float2 ea = shfl_xor1(v[j]);
    const float2 eb = shfl_xor1(v[16 + j]);
    float2 a = p ? ea : v[j];
    float2 b = p ? eb : v[16 + j];
    float2 c = p ? v[j] : ea;
    float2 e = p ? v[16 + j] : eb;
    float2 w1, w2, w3;
    if (j == 0) {
      w1 = w2 = w3 = make_float2(1.f, 0.f);
    } else {
      w1 = tw_n[j];
      w2 = tw_n[2 * j];
      w3 = tw_n[3 * j];
    }
    bfly4<SIGN>(a, b, c, e, w1, w2, w3);
    v[j] = p ? c : a;
    v[16 + j] = p ? e : b;
  }
  float2* __restrict__ ocolbase = out + base;
#pragma unroll
  for (int r = 0; r < 32; ++r) {
    const int k = 32 * p + r;
    float2 rr = v[r];
    if constexpr (TWIDDLE) {
      const unsigned long long m_ =
          (q0 * d.tw_f0 * (unsigned long long)k) & d.tw_mask;
      rr = cmulf(rr, tw_eval(m_, d.tw_angle));
    }
    st_stream(ocolbase + (uint32_t)k * stride32, rr, d.tuning);
  }
}

// ---------------------------------------------------------------------------
// N = 32 column pass, two lanes per column (same construction as pair64):
// lane p holds post-σ positions 16p..16p+15, stages L=4 and L=16 are
// lane-local, the final radix-2 stage (L=32) exchanges halves with one
// shfl_xor(…,1) per butterfly.  16 data registers per lane → higher
// occupancy than the mono col<32> kernel (~143 VGPRs, 3 waves/SIMD),
// which matters most for the VALU-heavy fused-preop (fp64 dedispersion)
// backward pass.
// ---------------------------------------------------------------------------

constexpr int col_sigma_inv32(int pos) {
  for (int i = 0; i < 32; ++i)
    if (col_sigma<32>(i) == pos) return i;
  return 0;
}

template <bool TWIDDLE, int SIGN, bool PREOP, int DEC = 0>
__global__ void __launch_bounds__(256)
    k_fft_col_pair32(const float2* __restrict__ in, float2* __restrict__ out,
                     FftPassDescDev d, unsigned long long n_ffts,
                     const float2* __restrict__ tw_n,
                     const float2* __restrict__ tw_hi,
                     const float2* __restrict__ tw_lo, FftPreopDev pre,
                     const uint8_t* __restrict__ raw2) {
  const unsigned long long tid =
      (unsigned long long)xcd_swizzle(blockIdx.x, gridDim.x, d.tuning) *
          blockDim.x + threadIdx.x;
  const unsigned long long id = tid >> 1;  // column index
  const int p = (int)(tid & 1);            // which half of the column
  if (id >= n_ffts) return;
  unsigned long long q0, q1, q2;
  digits(id, d, q0, q1, q2);
  const unsigned long long base = q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
  float thr_mean = 0.f;
  if constexpr (PREOP) {
    if (pre.mean_power) thr_mean = pre.threshold * (float)(*pre.mean_power);
  }
  float2 v[16];
  const float2* __restrict__ colbase = in + base;
  const uint32_t stride32 = (uint32_t)d.in_stride;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int i = p ? col_sigma_inv32(16 + r) : col_sigma_inv32(r);
    const uint32_t off = (uint32_t)i * stride32;
    float2 x;
    if constexpr (DEC != 0) x = decN_load<DEC>(raw2, base + off);
    else x = ld_stream(colbase + off, d.tuning);
    if constexpr (PREOP) {
      const unsigned long long flat = base + off;
      if (pre.r2c_m) x = r2c_combine_load(in, x, flat, pre.r2c_m);
      bool zap = pre.mean_power && (norm2(x) > thr_mean);
      for (int z = 0; z < pre.n_zap; ++z)
        zap |= (flat >= pre.zap[z].lo) & (flat <= pre.zap[z].hi);
      if (zap) {
        x = make_float2(0.f, 0.f);
      } else {
        if (pre.mean_power) {
          x.x *= pre.norm_coeff;
          x.y *= pre.norm_coeff;
        }
        x = cmulf(x, pre.table ? pre.table[flat]
                               : srtb_dedisp_factor(flat, pre.f_min, pre.f_c,
                                                    pre.df, pre.dm));
      }
    }
    v[r] = x;
  }
  // stages 1-2 lane-local; twiddle scale of the FULL length-32 transform
  col_stage<16, 4, 4, SIGN, 8>(v, tw_n);
  col_stage<16, 16, 4, SIGN, 2>(v, tw_n);
  // final radix-2 stage (L=32, M=16, TS=1): pairs {j, j+16}
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    const float2 other = shfl_xor1(v[j]);
    const float2 a = p ? other : v[j];
    const float2 b = p ? v[j] : other;
    const float2 w = (j == 0) ? make_float2(1.f, 0.f) : tw_n[j];
    const float2 y = cmulf(b, w);
    v[j] = p ? make_float2(a.x - y.x, a.y - y.y)
             : make_float2(a.x + y.x, a.y + y.y);
  }
  float2* __restrict__ ocolbase = out + base;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int k = 16 * p + r;
    float2 rr = v[r];
    if constexpr (TWIDDLE) {
      const unsigned long long m_ =
          (q0 * d.tw_f0 * (unsigned long long)k) & d.tw_mask;
      rr = cmulf(rr, tw_eval(m_, d.tw_angle));
    }
    st_stream(ocolbase + (uint32_t)k * stride32, rr, d.tuning);
  }
}

// ---------------------------------------------------------------------------
// Final pass: in-place radix-4 DIF in LDS (single buffer — twice the F of the
// ping-pong kernel) with the base-4 digit reversal folded into the STORE
// index and the multi-digit output scatter (validated: dif_r4 prototype).
// ---------------------------------------------------------------------------

struct DifFinalDescDev {
  uint32_t n;  // pure 4^t
  int n_log2, f_log2;
  int j_bits;                        // log2(L/n) = total prefix bits
  unsigned long long row_len;        // L (batch row stride, elements)
  unsigned long long out_elem_coef;  // = L/n (output digit coefficient)
  int n_prefix;                      // prefix digit count (id low bits = k_0)
  int pf_bits[4];                    // log2 f_w, w = 0..n_prefix-1
  unsigned long long pf_coef[4];     // S_w = prod(f_{w+1}..f_{m-2})
};

// instance id = row * 2^j_bits + J, J = sum k_w * prod(f_0..f_{w-1});
// input block index P = sum k_w * S_w (stored layout); output base = row*L + J
__device__ inline void dif_addr(unsigned long long id,
                                const DifFinalDescDev& d,
                                unsigned long long& in_blk,
                                unsigned long long& obase) {
  const unsigned long long J = id & ((1ull << d.j_bits) - 1);
  const unsigned long long row = id >> d.j_bits;
  unsigned long long x = J, P = 0;
#pragma unroll
  for (int w = 0; w < 4; ++w) {
    if (w < d.n_prefix) {
      const unsigned long long dig = x & ((1ull << d.pf_bits[w]) - 1);
      P += dig * d.pf_coef[w];
      x >>= d.pf_bits[w];
    }
  }
  in_blk = row * d.row_len + P * d.n;  // element offset of the input block
  obase = row * d.row_len + J;
}

// LDS bank swizzle for the DIF data array: butterfly strides of 4^k element
// granularity put 2-4 lanes of a 32-lane b64 group on one bank (measured
// 4e8 conflicts/dispatch); e ^ ((e>>2)&31) is bijective within any pow2 row
// >= 32 and brute-force conflict-free for every stage pattern of n=256/1024.
__device__ inline int dif_swz(int i) { return i ^ ((i >> 2) & 31); }

__device__ inline unsigned rev4_bits(unsigned k, int t2 /* log2(n) */) {
  // base-4 digit reversal = bit reversal then swap adjacent bit pairs
  unsigned r = __brev(k) >> (32 - t2);
  return ((r & 0x55555555u) << 1) | ((r & 0xAAAAAAAAu) >> 1);
}

template <int SIGN, bool SK>
__global__ void __launch_bounds__(256)
    k_fft_dif_final(const float2* __restrict__ in, float2* __restrict__ out,
                    DifFinalDescDev d, const float2* __restrict__ tw_n,
                    float2* __restrict__ sk_partials, int wgs_per_row) {
  extern __shared__ float2 lds[];
  const int n = d.n;
  const int nl = d.n_log2;
  const int F = 1 << d.f_log2;
  const int ldst = n + 2;
  float2* ltw = lds;
  float2* X = lds + n;
  const unsigned long long fft0 = (unsigned long long)blockIdx.x << d.f_log2;
  const int total = F << nl;

  for (int j = threadIdx.x; j < n; j += blockDim.x) ltw[j] = tw_n[j];

  // ---- contiguous load (register-staged chunks of 8) ----
  {
    const int iters = total >> 8;
    int it = 0;
    for (; it + 8 <= iters; it += 8) {
      float2 tmp[8];
      int lidx[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const int e = ((it + k) << 8) + threadIdx.x;
        const int f = e >> nl;
        const int i = e & (n - 1);
        unsigned long long in_blk, obase;
        dif_addr(fft0 + f, d, in_blk, obase);
        tmp[k] = in[in_blk + i];
        lidx[k] = f * ldst + dif_swz(i);
      }
#pragma unroll
      for (int k = 0; k < 8; ++k) X[lidx[k]] = tmp[k];
    }
    for (int e = (it << 8) + threadIdx.x; e < total; e += blockDim.x) {
      const int f = e >> nl;
      const int i = e & (n - 1);
      unsigned long long in_blk, obase;
      dif_addr(fft0 + f, d, in_blk, obase);
      X[f * ldst + dif_swz(i)] = in[in_blk + i];
    }
  }
  __syncthreads();

  // ---- in-place radix-4 DIF stages ----
  const int quarter = n >> 2;
  for (int L = n, tl = 0; L >= 4; L >>= 2, tl += 2) {
    const int M = L >> 2;
    const int m_log2 = (nl - tl) - 2;  // log2(M)
    for (int b = threadIdx.x; b < F * quarter; b += blockDim.x) {
      const int f = b >> (nl - 2);
      const int bb = b & (quarter - 1);
      // bb -> (block g, j): j = bb & (M-1); g = (bb >> m_log2) * L
      const int j = bb & (M - 1);
      const int g = (bb >> m_log2) << (m_log2 + 2);
      float2* row = X + f * ldst;
      const int e0 = dif_swz(g + j);
      const int e1 = dif_swz(g + j + M);
      const int e2 = dif_swz(g + j + 2 * M);
      const int e3 = dif_swz(g + j + 3 * M);
      const float2 a = row[e0];
      const float2 bv = row[e1];
      const float2 c = row[e2];
      const float2 dv = row[e3];
      // DFT over t then post-twiddle W_L^{m*j} (dif_r4 prototype)
      const float2 t0 = make_float2(a.x + c.x, a.y + c.y);
      const float2 t1 = make_float2(a.x - c.x, a.y - c.y);
      const float2 t2 = make_float2(bv.x + dv.x, bv.y + dv.y);
      const float2 dmy = make_float2(bv.x - dv.x, bv.y - dv.y);
      const float2 t3 = (SIGN > 0) ? make_float2(-dmy.y, dmy.x)
                                   : make_float2(dmy.y, -dmy.x);
      const float2 x0 = make_float2(t0.x + t2.x, t0.y + t2.y);
      const float2 x1 = make_float2(t1.x + t3.x, t1.y + t3.y);
      const float2 x2 = make_float2(t0.x - t2.x, t0.y - t2.y);
      const float2 x3 = make_float2(t1.x - t3.x, t1.y - t3.y);
      const int ts = tl;  // W_L^x = ltw[x << tl]
      row[e0] = x0;
      row[e1] = (j == 0) ? x1 : cmulf(x1, ltw[(j << ts)]);
      row[e2] = (j == 0) ? x2 : cmulf(x2, ltw[(2 * j) << ts]);
      row[e3] = (j == 0) ? x3 : cmulf(x3, ltw[(3 * j) << ts]);
    }
    __syncthreads();
  }

  // ---- store: digit-reversed LDS read, contiguous-lane output runs ----
  float sk2 = 0.0f, sk4 = 0.0f;
  for (int e = threadIdx.x; e < total; e += blockDim.x) {
    const int f = e & (F - 1);
    const int k = e >> d.f_log2;
    unsigned long long in_blk, obase;
    dif_addr(fft0 + f, d, in_blk, obase);
    const unsigned rk = rev4_bits((unsigned)k, nl);
    const float2 v = X[f * ldst + dif_swz((int)rk)];
    out[obase + (unsigned long long)k * d.out_elem_coef] = v;
    if constexpr (SK) {
      const float p = v.x * v.x + v.y * v.y;
      sk2 += p;
      sk4 += p * p;
    }
  }
  if constexpr (SK) {
    // all F instances of this WG belong to one batch row (the planner
    // guarantees (L/n) % F == 0); deterministic per-WG partial, no atomics
    const float b2 = block_reduce_sum(sk2);
    const float b4 = block_reduce_sum(sk4);
    if (threadIdx.x == 0) {
      const unsigned long long row = fft0 >> d.j_bits;
      const unsigned long long wg_in_row =
          (fft0 & ((1ull << d.j_bits) - 1)) >> d.f_log2;
      sk_partials[row * wgs_per_row + wg_in_row] = make_float2(b2, b4);
    }
  }
}

__global__ void k_sk_combine_partials(const float2* __restrict__ partials,
                                      size_t rows, int wgs_per_row,
                                      float2* __restrict__ s2s4) {
  const size_t r = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (r >= rows) return;
  float s2 = 0.0f, s4 = 0.0f;
  for (int w = 0; w < wgs_per_row; ++w) {
    const float2 p = partials[r * wgs_per_row + w];
    s2 += p.x;
    s4 += p.y;
  }
  s2s4[r] = make_float2(s2, s4);
}

// twiddle-table builder (fp64 on device)
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

// ---------------------------------------------------------------------------
// Wave-local small FFT (n = 64*E, E in {4,8,16} -> n in {256,512,1024}):
// ONE FFT per wave.  Per-lane E-point register FFT over the stride-64
// dimension, inter-step twiddle W_n^(lane*k2), then a 64-point DIF across
// lanes via shfl_xor butterflies (bit-reversed lane output), finished by a
// padded-row LDS transpose so the global store is coalesced.  No
// __syncthreads anywhere on the data path (wave-synchronous), one LDS
// round trip per element vs the ping-pong Stockham's log4(n).
// Index math validated in scripts/wavefft_proto.py.
// ---------------------------------------------------------------------------

__device__ inline float2 shfl_xor_f2(float2 x, int mask) {
  return make_float2(__shfl_xor(x.x, mask, 64), __shfl_xor(x.y, mask, 64));
}

template <int SIGN, int E, bool GT = false>
__global__ void __launch_bounds__(256)
    k_fft_wave(const float2* __restrict__ in, float2* __restrict__ out,
               unsigned long long n_ffts, const float2* __restrict__ tw_n) {
  constexpr int N = 64 * E;
  constexpr int EL2 = col_ilog2(E);
  extern __shared__ float2 lds[];
  // layout: [ltw: N][w64: 64][per-wave transpose buffers: 4 * 64*(E+1)];
  // GT = twiddles read from GLOBAL (L2-hot) — frees N+64 float2 of LDS
  // for one more workgroup per CU
  const int wave = (int)(threadIdx.x >> 6);
  const int lane = (int)(threadIdx.x & 63);
  const float2* ltw;
  const float2* w64;
  float2* buf;
  if constexpr (GT) {
    ltw = tw_n;
    w64 = nullptr;
    buf = lds + (size_t)wave * 64 * (E + 1);
  } else {
    float2* ltw_l = lds;
    float2* w64_l = lds + N;
    for (int j = (int)threadIdx.x; j < N; j += (int)blockDim.x)
      ltw_l[j] = tw_n[j];
    if (threadIdx.x < 64) w64_l[threadIdx.x] = tw_n[(size_t)threadIdx.x * E];
    __syncthreads();  // tables only; the data path below is wave-local
    ltw = ltw_l;
    w64 = w64_l;
    buf = w64_l + 64 + (size_t)wave * 64 * (E + 1);
  }

  const unsigned rev = (__brev((unsigned)lane) >> 26);  // bitrev6(lane)
  const unsigned long long wave0 =
      (unsigned long long)blockIdx.x * (blockDim.x >> 6) + wave;
  const unsigned long long wstride =
      (unsigned long long)gridDim.x * (blockDim.x >> 6);

  for (unsigned long long r = wave0; r < n_ffts; r += wstride) {
    const float2* __restrict__ src = in + r * N;
    float2* __restrict__ dst = out + r * N;
    // step 1: per-lane E-point FFT of src[lane + 64*i2] (sigma-permuted
    // load, col_fft needs the E-entry table = ltw strided by 64... the
    // E-point twiddles are W_E^j = W_N^(64 j) = ltw[64*j])
    float2 v[E];
#pragma unroll
    for (int i2 = 0; i2 < E; ++i2)
      v[col_sigma<E>(i2)] = src[lane + 64 * i2];
    {
      // col_fft expects an E-entry table at stride 1: gather through a
      // small register-resident view is not possible, so give it a lambda-
      // free path: reuse ltw with TS scaling by calling the stages directly
      // (col_stage uses tw[j*TS]; our table is ltw with W_E^j at 64*j, so
      // scale TS by 64)
      if constexpr (E >= 4) col_stage<E, 4, 4, SIGN, 16 * E>(v, ltw);
      if constexpr (E >= 16) col_stage<E, 16, 4, SIGN, 4 * E>(v, ltw);
      if constexpr (E == 8) col_stage<E, 8, 2, SIGN, 8 * E>(v, ltw);
      if constexpr (E == 32) col_stage<E, 32, 2, SIGN, 2 * E>(v, ltw);
    }
    // step 2: twiddle W_N^(lane*k2) by recurrence from W_N^lane — the
    // direct gather ltw[lane*k2] is a stride-k2 LDS access (up to 8-way
    // bank conflicts at even k2); one stride-1 gather + E-2 register
    // cmuls is conflict-free (phase error ~E ulp, within fp32 FFT noise)
    {
      const float2 wbase = ltw[lane];
      float2 w = wbase;
#pragma unroll
      for (int k2 = 1; k2 < E; ++k2) {
        v[k2] = cmulf(v[k2], w);
        w = cmulf(w, wbase);
      }
    }
    // step 3: 64-point DIF across lanes; output lane holds k1 = rev
#pragma unroll
    for (int M = 32; M >= 1; M >>= 1) {
      const int upper = lane & M;
      const int j = lane & (M - 1);
      const float2 tw = GT ? tw_n[(size_t)(j * (32 / M)) * E]
                           : w64[j * (32 / M)];
#pragma unroll
      for (int k2 = 0; k2 < E; ++k2) {
        const float2 t = shfl_xor_f2(v[k2], M);
        if (upper) {
          v[k2] = cmulf(make_float2(t.x - v[k2].x, t.y - v[k2].y), tw);
        } else {
          v[k2] = make_float2(v[k2].x + t.x, v[k2].y + t.y);
        }
      }
    }
    // step 4: padded-row LDS transpose (row = k1, stride E+1) -> coalesced
    // store.  Wave-local: the compiler's lgkmcnt waits order write/read.
#pragma unroll
    for (int k2 = 0; k2 < E; ++k2) buf[rev * (E + 1) + k2] = v[k2];
    __builtin_amdgcn_wave_barrier();
#pragma unroll
    for (int t = 0; t < E; ++t) {
      const int k = lane + 64 * t;
      dst[k] = buf[(k >> EL2) * (E + 1) + (k & (E - 1))];
    }
    __builtin_amdgcn_wave_barrier();
  }
}


// r2c post-process (packed-real trick; fftref.r2c_post):
//   E = (Z[k]+conj(Z[M-k]))/2,  O = -i/2*(Z[k]-conj(Z[M-k])),
//   w(k) = exp(-2πi k/(2M)),
//   X[k] = E + w*O  and  X[M-k] = conj(E - w*O),
// each thread handles the (k, M-k) PAIR so the kernel is in-place safe
// (x may alias z).  k=0: X[0]=Re(Z0)+Im(Z0) (Nyquist dropped, count = Nc).
// Optionally accumulates Σ|X|² partials (fused RFI-s1 mean-power — saves a
// full 4 GB spectrum read).
// ---------------------------------------------------------------------------
// EXPERIMENTAL (round-2 lever, reachable only via SRTB_FFT_FACTORS):
// strided middle pass of length 512 through LDS — replaces TWO register-
// column passes, cutting a whole 8 GB sweep from the 2^29 plan
// ([64,64,512]+256 = 4 passes vs [64,64,64,8]+256 = 5).
//
// 512 = 2·256: stage A is the radix-2 DIF split (odd half twiddled by
// W512^i), then two independent in-place radix-4 DIF-256 per column (the
// validated k_fft_dif_final butterflies + swizzle), and the store reads
// LDS at 256·(k&1) + swz(rev4(k>>1)) so output lands in natural order.
// F = 32 columns per workgroup → every global access instruction touches
// 32 consecutive columns (256 B runs at any stride).  LDS: 32 rows of
// 513 float2 (row stride 1026 dwords ≡ 2 mod 64 → cross-column accesses
// are bank-conflict-free).  Math validated against numpy (oracle sim).
// ---------------------------------------------------------------------------

template <int SIGN, int F = 32>
__global__ void __launch_bounds__(256)
    k_fft_mid512(const float2* __restrict__ in, float2* __restrict__ out,
                 FftPassDescDev d, unsigned long long n_ffts,
                 const float2* __restrict__ tw_n) {
  extern __shared__ float2 lds[];
  constexpr int LOGF = col_ilog2(F);  // columns per workgroup
  constexpr int LDST = 513;   // row stride (float2)
  float2* ltw = lds;          // 512-entry twiddle table
  float2* X = lds + 512;
  const unsigned long long col0 = (unsigned long long)blockIdx.x * F;

  for (int j = threadIdx.x; j < 512; j += blockDim.x) ltw[j] = tw_n[j];

  // ---- load: lane f fastest → F consecutive columns per instruction ----
  unsigned long long base_f;  // this lane's column base (f = tid & (F-1))
  {
    const int f = threadIdx.x & (F - 1);
    unsigned long long q0, q1, q2;
    digits(col0 + f, d, q0, q1, q2);
    base_f = q0 * d.in_c0 + q1 * d.in_c1 + q2 * d.in_c2;
  }
  __syncthreads();
  const uint32_t stride32 = (uint32_t)d.in_stride;
  {
    const int f = threadIdx.x & (F - 1);
    const bool active = (col0 + f) < n_ffts;
    constexpr int TPC = 256 / F;  // i-slices per thread pass
    for (int r = 0; r < 512 / TPC; ++r) {
      const int i = r * TPC + (threadIdx.x >> LOGF);  // i in [0,512)
      const int h = i >> 8, j = i & 255;
      X[f * LDST + 256 * h + dif_swz(j)] =
          active ? in[base_f + (uint32_t)i * stride32]
                 : make_float2(0.f, 0.f);
    }
  }
  __syncthreads();

  // ---- stage A: radix-2 split, odd half × W512^i ----
  for (int b = threadIdx.x; b < F * 256; b += blockDim.x) {
    const int f = b & (F - 1);
    const int i = b >> LOGF;
    float2* row = X + f * LDST;
    const int e = dif_swz(i);
    const float2 u = row[e];
    const float2 v = row[256 + e];
    row[e] = make_float2(u.x + v.x, u.y + v.y);
    const float2 dmy = make_float2(u.x - v.x, u.y - v.y);
    row[256 + e] = (i == 0) ? dmy : cmulf(dmy, ltw[i]);
  }
  __syncthreads();

  // ---- stage B: two independent in-place radix-4 DIF-256 per column ----
  // W_L^x within a 256-point half = tw512[x * 512/L] = ltw[x << (9-log2 L)]
  for (int L = 256, tl = 1; L >= 4; L >>= 2, tl += 2) {
    const int M = L >> 2;
    const int m_log2 = __builtin_ctz((unsigned)M);
    for (int b = threadIdx.x; b < F * 128; b += blockDim.x) {
      const int f = b & (F - 1);
      const int rest = b >> LOGF;     // [0,128): quartet index + half
      const int h = rest >> 6;
      const int bb = rest & 63;
      const int j = bb & (M - 1);
      const int g = (bb >> m_log2) << (m_log2 + 2);
      float2* row = X + f * LDST + 256 * h;
      const int e0 = dif_swz(g + j);
      const int e1 = dif_swz(g + j + M);
      const int e2 = dif_swz(g + j + 2 * M);
      const int e3 = dif_swz(g + j + 3 * M);
      const float2 a = row[e0];
      const float2 bv = row[e1];
      const float2 c = row[e2];
      const float2 dv = row[e3];
      const float2 t0 = make_float2(a.x + c.x, a.y + c.y);
      const float2 t1 = make_float2(a.x - c.x, a.y - c.y);
      const float2 t2 = make_float2(bv.x + dv.x, bv.y + dv.y);
      const float2 dmy = make_float2(bv.x - dv.x, bv.y - dv.y);
      const float2 t3 = (SIGN > 0) ? make_float2(-dmy.y, dmy.x)
                                   : make_float2(dmy.y, -dmy.x);
      const float2 x0 = make_float2(t0.x + t2.x, t0.y + t2.y);
      const float2 x1 = make_float2(t1.x + t3.x, t1.y + t3.y);
      const float2 x2 = make_float2(t0.x - t2.x, t0.y - t2.y);
      const float2 x3 = make_float2(t1.x - t3.x, t1.y - t3.y);
      row[e0] = x0;
      row[e1] = (j == 0) ? x1 : cmulf(x1, ltw[j << tl]);
      row[e2] = (j == 0) ? x2 : cmulf(x2, ltw[(2 * j) << tl]);
      row[e3] = (j == 0) ? x3 : cmulf(x3, ltw[(3 * j) << tl]);
    }
    __syncthreads();
  }

  // ---- store: natural k, inter-pass twiddle W^(q0*f0*k) ----
  {
    const int f = threadIdx.x & (F - 1);
    unsigned long long q0, q1, q2;
    digits(col0 + f, d, q0, q1, q2);
    const bool active = (col0 + f) < n_ffts;
    constexpr int TPC = 256 / F;
    for (int r = 0; r < 512 / TPC; ++r) {
      const int k = r * TPC + (threadIdx.x >> LOGF);
      const unsigned rk = rev4_bits((unsigned)(k >> 1), 8);
      const float2 v = X[f * LDST + 256 * (k & 1) + dif_swz((int)rk)];
      float2 rr = v;
      const unsigned long long m_ =
          (q0 * d.tw_f0 * (unsigned long long)k) & d.tw_mask;
      rr = cmulf(rr, tw_eval(m_, d.tw_angle));
      if (active) out[base_f + (uint32_t)k * stride32] = rr;
    }
  }
}

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
    const float2 dif = make_float2(zk.x - zmc.x, zk.y - zmc.y);
    const float2 odd = make_float2(0.5f * dif.y, -0.5f * dif.x);
    // argument reduced in fp64 (k/m exact), fast f32 sincos (~1.5e-6 rad)
    const float phiw = (float)(-M_PI * (double)k / (double)m);
    float sw, cw;
    __sincosf(phiw, &sw, &cw);
    const float2 w = make_float2(cw, sw);
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

inline int ilog2(unsigned long long v) {
  int t = 0;
  while ((1ull << t) < v) ++t;
  return t;
}

}  // namespace

hipError_t fft_build_twiddle(float2* table, size_t count, double m, int sign,
                             hipStream_t stream) {
  hipLaunchKernelGGL(k_build_twiddle, grid_for(count), dim3(kBlock), 0,
                     stream, table, count, sign * 2.0 * M_PI / m);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

static bool use_pair32() {
  // default ON since the round-2 soak (300 iters on the flagship backward
  // shape, bit-correct, 5.79 vs 5.92 ms mono; the round-1 memory fault
  // never reproduced); SRTB_FFT_PAIR32=0 falls back to the mono kernel
  static const bool v = [] {
    const char* e = std::getenv("SRTB_FFT_PAIR32");
    return e ? (std::atoi(e) != 0) : true;
  }();
  return v;
}

// column-pass tuning experiments: SRTB_FFT_SWIZZLE=1 (XCD-aware workgroup
// remap), SRTB_FFT_NT=1 (non-temporal loads/stores)
static uint32_t fft_tuning() {
  static const uint32_t v = [] {
    uint32_t t = 0;
    if (const char* e = std::getenv("SRTB_FFT_SWIZZLE"))
      if (std::atoi(e)) t |= 1u;
    if (const char* e = std::getenv("SRTB_FFT_NT"))
      if (std::atoi(e)) t |= 2u;
    // bit2: radix-16 Stockham stages — measured 4-5% SLOWER than radix-4
    // at every single-pass length (r02 check4: 0.724/0.810/1.468 vs
    // 0.691/0.773/1.435 ms at 256/1024/2048) despite 5->3 LDS round
    // trips; the per-butterfly 15-gather inter-stage twiddle reads eat
    // the barrier savings.  Kept opt-in for future work.
    if (const char* e = std::getenv("SRTB_FFT_SP16"))
      if (std::atoi(e)) t |= 4u;
    return t;
  }();
  return v;
}

hipError_t fft_stockham_pass(const float2* in, float2* out,
                             const FftPassDesc& hd, size_t n_ffts, int F,
                             bool load_ffast, bool store_ffast, int sign,
                             const float2* tw_n, const float2* tw_hi,
                             const float2* tw_lo, hipStream_t stream) {
  if ((hd.n & (hd.n - 1)) || (F & (F - 1))) return hipErrorInvalidValue;
  if (hd.d0 & (hd.d0 - 1)) return hipErrorInvalidValue;
  if (hd.d1 & (hd.d1 - 1)) return hipErrorInvalidValue;
  FftPassDescDev d;
  d.n = hd.n;
  d.n_log2 = ilog2(hd.n);
  d.f_log2 = ilog2((unsigned)F);
  // d0 == 0: contiguous-rows sentinel -> q0 = id (48-bit mask), q1 = q2 = 0
  d.d0_log2 = hd.d0 ? ilog2(hd.d0) : 48;
  d.d1_log2 = ilog2(hd.d1 ? hd.d1 : 1);
  d.in_c0 = hd.in_c0; d.in_c1 = hd.in_c1; d.in_c2 = hd.in_c2;
  d.in_stride = hd.in_stride;
  d.out_c0 = hd.out_c0; d.out_c1 = hd.out_c1; d.out_c2 = hd.out_c2;
  d.out_stride = hd.out_stride;
  d.tw_f0 = hd.tw_f0; d.tw_f1 = hd.tw_f1;
  d.tw_mask = hd.tw_mod ? hd.tw_mod - 1 : 0;
  d.tw_lo_bits = hd.tw_lo_bits;
  d.tw_angle = hd.tw_angle;
  const bool twiddle = hd.tw_mod != 0;
  if (n_ffts % F != 0) return hipErrorInvalidValue;
  d.tuning = fft_tuning();
  const uint32_t grid = (uint32_t)(n_ffts / F);
  const size_t lds_bytes =
      ((size_t)hd.n + 2ull * F * (hd.n + 2) + 16) * sizeof(float2);
  if (lds_bytes > 160 * 1024) return hipErrorInvalidValue;

#define DISPATCH4(LF, SF, TW, SG)                                           \
  hipLaunchKernelGGL((k_fft_stockham<LF, SF, TW, SG>), dim3(grid),          \
                     dim3(256), lds_bytes, stream, in, out, d, tw_n, tw_hi, \
                     tw_lo)
#define DISPATCH2(LF, SF)                                                   \
  do {                                                                      \
    if (twiddle) {                                                          \
      if (sign < 0) DISPATCH4(LF, SF, true, -1);                            \
      else          DISPATCH4(LF, SF, true, 1);                             \
    } else {                                                                \
      if (sign < 0) DISPATCH4(LF, SF, false, -1);                           \
      else          DISPATCH4(LF, SF, false, 1);                            \
    }                                                                       \
  } while (0)
  if (load_ffast) {
    if (store_ffast) DISPATCH2(true, true);
    else             DISPATCH2(true, false);
  } else {
    if (store_ffast) DISPATCH2(false, true);
    else             DISPATCH2(false, false);
  }
#undef DISPATCH2
#undef DISPATCH4
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}


hipError_t fft_col_pass(const float2* in, float2* out, const FftPassDesc& hd,
                        size_t n_ffts, int sign, const float2* tw_n,
                        const float2* tw_hi, const float2* tw_lo,
                        hipStream_t stream, const FftPreop* preop,
                        const uint8_t* raw2, int raw_bits) {
  if (raw2 && (preop || hd.tw_mod == 0)) return hipErrorInvalidValue;
  if (raw2 && raw_bits != 1 && raw_bits != 2 && raw_bits != 4 &&
      raw_bits != 8 && raw_bits != -8 && raw_bits != 16 && raw_bits != -16)
    return hipErrorInvalidValue;
  FftPreopDev pre{};
  if (preop) {
    pre.mean_power = preop->mean_power;
    pre.threshold = preop->threshold;
    pre.norm_coeff = preop->norm_coeff;
    pre.n_zap = preop->n_zap;
    for (int i = 0; i < preop->n_zap; ++i) pre.zap[i] = preop->zap[i];
    pre.f_min = preop->f_min;
    pre.f_c = preop->f_c;
    pre.df = preop->df;
    pre.dm = preop->dm;
    pre.table = preop->table;
    pre.r2c_m = preop->r2c_m;
  }
  FftPassDescDev d;
  d.n = hd.n;
  d.n_log2 = ilog2(hd.n);
  d.f_log2 = 0;
  d.d0_log2 = hd.d0 ? ilog2(hd.d0) : 48;
  d.d1_log2 = ilog2(hd.d1 ? hd.d1 : 1);
  d.in_c0 = hd.in_c0; d.in_c1 = hd.in_c1; d.in_c2 = hd.in_c2;
  d.in_stride = hd.in_stride;
  d.out_c0 = hd.out_c0; d.out_c1 = hd.out_c1; d.out_c2 = hd.out_c2;
  d.out_stride = hd.out_stride;
  d.tw_f0 = hd.tw_f0; d.tw_f1 = hd.tw_f1;
  d.tw_mask = hd.tw_mod ? hd.tw_mod - 1 : 0;
  d.tw_lo_bits = hd.tw_lo_bits;
  d.tw_angle = hd.tw_angle;
  d.tuning = fft_tuning();
  const bool twiddle = hd.tw_mod != 0;
  const uint32_t grid = (uint32_t)((n_ffts + 255) / 256);

#define COL_LAUNCH(N, TW, SG, PR)                                           \
  hipLaunchKernelGGL((k_fft_col<N, TW, SG, PR>), dim3(grid), dim3(256), 0,   \
                     stream, in, out, d, n_ffts, tw_n, tw_hi, tw_lo, pre,    \
                     raw2)
#define COL_LAUNCH_D(N, SG, B)                                               \
  hipLaunchKernelGGL((k_fft_col<N, true, SG, false, B>), dim3(grid),         \
                     dim3(256), 0, stream, in, out, d, n_ffts, tw_n, tw_hi,  \
                     tw_lo, pre, raw2)
#define COL_DISPATCH_D(N)                                                    \
  if (raw_bits == 1) { if (sign < 0) COL_LAUNCH_D(N, -1, 1);                 \
                       else COL_LAUNCH_D(N, 1, 1); }                         \
  else if (raw_bits == 2) { if (sign < 0) COL_LAUNCH_D(N, -1, 2);            \
                            else COL_LAUNCH_D(N, 1, 2); }                    \
  else if (raw_bits == 4) { if (sign < 0) COL_LAUNCH_D(N, -1, 4);            \
                            else COL_LAUNCH_D(N, 1, 4); }                    \
  else if (raw_bits == 8) { if (sign < 0) COL_LAUNCH_D(N, -1, 8);            \
                            else COL_LAUNCH_D(N, 1, 8); }                    \
  else if (raw_bits == -8) { if (sign < 0) COL_LAUNCH_D(N, -1, -8);          \
                             else COL_LAUNCH_D(N, 1, -8); }                  \
  else if (raw_bits == 16) { if (sign < 0) COL_LAUNCH_D(N, -1, 16);          \
                             else COL_LAUNCH_D(N, 1, 16); }                  \
  else { if (sign < 0) COL_LAUNCH_D(N, -1, -16);                             \
         else COL_LAUNCH_D(N, 1, -16); }
#define COL_DISPATCH(N)                                                      \
  case N:                                                                    \
    if (raw2) {                                                              \
      COL_DISPATCH_D(N)                                                      \
    } else if (twiddle) {                                                           \
      if (sign < 0) { if (preop) COL_LAUNCH(N, true, -1, true);              \
                      else COL_LAUNCH(N, true, -1, false); }                 \
      else          { if (preop) COL_LAUNCH(N, true, 1, true);               \
                      else COL_LAUNCH(N, true, 1, false); }                  \
    } else {                                                                 \
      if (sign < 0) { if (preop) COL_LAUNCH(N, false, -1, true);             \
                      else COL_LAUNCH(N, false, -1, false); }                \
      else          { if (preop) COL_LAUNCH(N, false, 1, true);              \
                      else COL_LAUNCH(N, false, 1, false); }                 \
    }                                                                        \
    break;
  // N = 64 uses the lane-pair kernel (two threads per column)
  const uint32_t grid2 = (uint32_t)((2 * n_ffts + 255) / 256);
#define COL_LAUNCH_P(TW, SG, PR)                                             \
  hipLaunchKernelGGL((k_fft_col_pair64<TW, SG, PR>), dim3(grid2), dim3(256), \
                     0, stream, in, out, d, n_ffts, tw_n, tw_hi, tw_lo, pre, \
                     raw2)
#define COL_LAUNCH_PD(SG, B)                                                 \
  hipLaunchKernelGGL((k_fft_col_pair64<true, SG, false, B>), dim3(grid2),    \
                     dim3(256), 0, stream, in, out, d, n_ffts, tw_n, tw_hi,  \
                     tw_lo, pre, raw2)
#define COL_LAUNCH_P32(TW, SG, PR)                                           \
  hipLaunchKernelGGL((k_fft_col_pair32<TW, SG, PR>), dim3(grid2), dim3(256), \
                     0, stream, in, out, d, n_ffts, tw_n, tw_hi, tw_lo, pre, \
                     raw2)
#define COL_LAUNCH_P32D(SG, B)                                               \
  hipLaunchKernelGGL((k_fft_col_pair32<true, SG, false, B>), dim3(grid2),    \
                     dim3(256), 0, stream, in, out, d, n_ffts, tw_n, tw_hi,  \
                     tw_lo, pre, raw2)
  switch (hd.n) {
    COL_DISPATCH(2)
    COL_DISPATCH(4)
    COL_DISPATCH(8)
    COL_DISPATCH(16)
    // N = 32: the lane-pair kernel (6 waves/SIMD) measured end-to-end
    // neutral vs the mono kernel (3 waves) and is opt-in via
    // SRTB_FFT_PAIR32=1 pending more soak coverage; mono is the default.
    case 32:
      if (use_pair32()) {
        if (raw2) {
          if (raw_bits == 1) { if (sign < 0) COL_LAUNCH_P32D(-1, 1);
                               else COL_LAUNCH_P32D(1, 1); }
          else if (raw_bits == 2) { if (sign < 0) COL_LAUNCH_P32D(-1, 2);
                                    else COL_LAUNCH_P32D(1, 2); }
          else if (raw_bits == 4) { if (sign < 0) COL_LAUNCH_P32D(-1, 4);
                                    else COL_LAUNCH_P32D(1, 4); }
          else if (raw_bits == 8) { if (sign < 0) COL_LAUNCH_P32D(-1, 8);
                                    else COL_LAUNCH_P32D(1, 8); }
          else if (raw_bits == -8) { if (sign < 0) COL_LAUNCH_P32D(-1, -8);
                                     else COL_LAUNCH_P32D(1, -8); }
          else if (raw_bits == 16) { if (sign < 0) COL_LAUNCH_P32D(-1, 16);
                                     else COL_LAUNCH_P32D(1, 16); }
          else { if (sign < 0) COL_LAUNCH_P32D(-1, -16);
                 else COL_LAUNCH_P32D(1, -16); }
          break;
        }
        if (twiddle) {
          if (sign < 0) { if (preop) COL_LAUNCH_P32(true, -1, true);
                          else COL_LAUNCH_P32(true, -1, false); }
          else          { if (preop) COL_LAUNCH_P32(true, 1, true);
                          else COL_LAUNCH_P32(true, 1, false); }
        } else {
          if (sign < 0) { if (preop) COL_LAUNCH_P32(false, -1, true);
                          else COL_LAUNCH_P32(false, -1, false); }
          else          { if (preop) COL_LAUNCH_P32(false, 1, true);
                          else COL_LAUNCH_P32(false, 1, false); }
        }
        break;
      }
      if (raw2) {
        COL_DISPATCH_D(32)
        break;
      }
      if (twiddle) {
        if (sign < 0) { if (preop) COL_LAUNCH(32, true, -1, true);
                        else COL_LAUNCH(32, true, -1, false); }
        else          { if (preop) COL_LAUNCH(32, true, 1, true);
                        else COL_LAUNCH(32, true, 1, false); }
      } else {
        if (sign < 0) { if (preop) COL_LAUNCH(32, false, -1, true);
                        else COL_LAUNCH(32, false, -1, false); }
        else          { if (preop) COL_LAUNCH(32, false, 1, true);
                        else COL_LAUNCH(32, false, 1, false); }
      }
      break;
    case 64:
      if (raw2) {
        if (raw_bits == 1) { if (sign < 0) COL_LAUNCH_PD(-1, 1);
                             else COL_LAUNCH_PD(1, 1); }
        else if (raw_bits == 2) { if (sign < 0) COL_LAUNCH_PD(-1, 2);
                                  else COL_LAUNCH_PD(1, 2); }
        else if (raw_bits == 4) { if (sign < 0) COL_LAUNCH_PD(-1, 4);
                                  else COL_LAUNCH_PD(1, 4); }
        else if (raw_bits == 8) { if (sign < 0) COL_LAUNCH_PD(-1, 8);
                                  else COL_LAUNCH_PD(1, 8); }
        else if (raw_bits == -8) { if (sign < 0) COL_LAUNCH_PD(-1, -8);
                                   else COL_LAUNCH_PD(1, -8); }
        else if (raw_bits == 16) { if (sign < 0) COL_LAUNCH_PD(-1, 16);
                                   else COL_LAUNCH_PD(1, 16); }
        else { if (sign < 0) COL_LAUNCH_PD(-1, -16);
               else COL_LAUNCH_PD(1, -16); }
        break;
      }
      if (twiddle) {
        if (sign < 0) { if (preop) COL_LAUNCH_P(true, -1, true);
                        else COL_LAUNCH_P(true, -1, false); }
        else          { if (preop) COL_LAUNCH_P(true, 1, true);
                        else COL_LAUNCH_P(true, 1, false); }
      } else {
        if (sign < 0) { if (preop) COL_LAUNCH_P(false, -1, true);
                        else COL_LAUNCH_P(false, -1, false); }
        else          { if (preop) COL_LAUNCH_P(false, 1, true);
                        else COL_LAUNCH_P(false, 1, false); }
      }
      break;
    default:
      return hipErrorInvalidValue;
  }
#undef COL_DISPATCH
#undef COL_DISPATCH_D
#undef COL_LAUNCH
#undef COL_LAUNCH_D
#undef COL_LAUNCH_P
#undef COL_LAUNCH_PD
#undef COL_LAUNCH_P32
#undef COL_LAUNCH_P32D
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t fft_mid512_pass(const float2* in, float2* out,
                           const FftPassDesc& hd, size_t n_ffts, int sign,
                           const float2* tw_n, hipStream_t stream) {
  if (hd.n != 512 || hd.tw_mod == 0) return hipErrorInvalidValue;
  FftPassDescDev d;
  d.n = hd.n;
  d.n_log2 = 9;
  d.f_log2 = 5;
  d.d0_log2 = hd.d0 ? ilog2(hd.d0) : 48;
  d.d1_log2 = ilog2(hd.d1 ? hd.d1 : 1);
  d.in_c0 = hd.in_c0; d.in_c1 = hd.in_c1; d.in_c2 = hd.in_c2;
  d.in_stride = hd.in_stride;
  d.out_c0 = hd.out_c0; d.out_c1 = hd.out_c1; d.out_c2 = hd.out_c2;
  d.out_stride = hd.out_stride;
  d.tw_f0 = hd.tw_f0; d.tw_f1 = hd.tw_f1;
  d.tw_mask = hd.tw_mod ? hd.tw_mod - 1 : 0;
  d.tw_lo_bits = hd.tw_lo_bits;
  d.tw_angle = hd.tw_angle;
  // SRTB_FFT_MIDF ∈ {8,16,32}: fewer columns per workgroup = less LDS =
  // more workgroups/CU (measured: occupancy, not run length, binds)
  static const int Fw = [] {
    const char* e = std::getenv("SRTB_FFT_MIDF");
    const int v = e ? std::atoi(e) : 32;
    return (v == 8 || v == 16) ? v : 32;
  }();
  const uint32_t grid = (uint32_t)((n_ffts + Fw - 1) / Fw);
  const size_t lds_bytes = (512 + (size_t)Fw * 513) * sizeof(float2);
  if (Fw == 8) {
    if (sign < 0)
      hipLaunchKernelGGL((k_fft_mid512<-1, 8>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
    else
      hipLaunchKernelGGL((k_fft_mid512<1, 8>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
  } else if (Fw == 16) {
    if (sign < 0)
      hipLaunchKernelGGL((k_fft_mid512<-1, 16>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
    else
      hipLaunchKernelGGL((k_fft_mid512<1, 16>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
  } else {
    if (sign < 0)
      hipLaunchKernelGGL((k_fft_mid512<-1, 32>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
    else
      hipLaunchKernelGGL((k_fft_mid512<1, 32>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, n_ffts, tw_n);
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t fft_wave_pass(const float2* in, float2* out, uint32_t n,
                         size_t n_ffts, int sign, const float2* tw_n,
                         hipStream_t stream) {
  if (n != 256 && n != 512 && n != 1024 && n != 2048)
    return hipErrorInvalidValue;
  const int E = (int)(n / 64);
  // SRTB_FFT_WAVE_GT=1: twiddles from GLOBAL (L2-hot) instead of LDS —
  // frees n+64 float2 of LDS for one more workgroup per CU (A/B probe)
  static const bool gt = [] {
    const char* e = std::getenv("SRTB_FFT_WAVE_GT");
    return e && std::atoi(e) != 0;
  }();
  const size_t lds_bytes =
      (gt ? 4ull * 64 * (E + 1)
          : (size_t)n + 64 + 4ull * 64 * (E + 1)) * sizeof(float2);
  // enough waves to fill the chip ~8x; each wave strides over FFTs
  uint32_t grid = (uint32_t)((n_ffts + 3) / 4);
  if (grid > 16384) grid = 16384;
#define WAVE_LAUNCH(SG, EE, G) hipLaunchKernelGGL((k_fft_wave<SG, EE, G>), dim3(grid), dim3(256), lds_bytes, stream, in, out, n_ffts, tw_n)
#define WAVE_DISPATCH(SG, EE) do { if (gt) WAVE_LAUNCH(SG, EE, true); else WAVE_LAUNCH(SG, EE, false); } while (0)
  if (sign < 0) {
    if (E == 4) WAVE_DISPATCH(-1, 4);
    else if (E == 8) WAVE_DISPATCH(-1, 8);
    else if (E == 16) WAVE_DISPATCH(-1, 16);
    else WAVE_DISPATCH(-1, 32);
  } else {
    if (E == 4) WAVE_DISPATCH(1, 4);
    else if (E == 8) WAVE_DISPATCH(1, 8);
    else if (E == 16) WAVE_DISPATCH(1, 16);
    else WAVE_DISPATCH(1, 32);
  }
#undef WAVE_DISPATCH
#undef WAVE_LAUNCH
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

namespace {
__global__ void k_r2c_mean_from_power(const float2* __restrict__ partials,
                                      size_t n_partials,
                                      const float2* __restrict__ z0, size_t m,
                                      double* __restrict__ out_mean) {
  double acc = 0.0;
  for (size_t i = threadIdx.x; i < n_partials; i += blockDim.x)
    acc += (double)partials[i].x;
  const double p = block_reduce_sum(acc);
  if (threadIdx.x == 0) {
    const float2 z = *z0;
    const double x0 = (double)z.x + z.y;
    const double xm = (double)z.x - z.y;
    *out_mean = (p + 0.5 * (x0 * x0 - xm * xm)) / (double)m;
  }
}
}  // namespace

hipError_t r2c_mean_from_power(const float2* power_partials,
                               size_t n_partials, const float2* z0, size_t m,
                               double* out_mean, hipStream_t stream) {
  hipLaunchKernelGGL(k_r2c_mean_from_power, dim3(1), dim3(256), 0, stream,
                     power_partials, n_partials, z0, m, out_mean);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t fft_dif_final(const float2* in, float2* out,
                         const DifFinalDesc& hd, size_t n_ffts, int F,
                         int sign, const float2* tw_n, float2* sk_partials,
                         hipStream_t stream) {
  if (hd.n & (hd.n - 1)) return hipErrorInvalidValue;
  DifFinalDescDev d;
  d.n = hd.n;
  d.n_log2 = ilog2(hd.n);
  if (d.n_log2 & 1) return hipErrorInvalidValue;  // pure radix-4 lengths only
  d.f_log2 = ilog2((unsigned)F);
  d.row_len = hd.out_c2;  // L
  d.out_elem_coef = hd.out_elem_coef;
  d.n_prefix = hd.n_prefix;
  int jb = 0;
  for (int i = 0; i < 4; ++i) {
    d.pf_bits[i] = hd.pf_bits[i];
    d.pf_coef[i] = hd.pf_coef[i];
    if (i < hd.n_prefix) jb += hd.pf_bits[i];
  }
  d.j_bits = jb;
  if (n_ffts % F != 0) return hipErrorInvalidValue;
  const uint32_t grid = (uint32_t)(n_ffts / F);
  const size_t lds_bytes =
      ((size_t)hd.n + (size_t)F * (hd.n + 2)) * sizeof(float2);
  if (lds_bytes > 160 * 1024) return hipErrorInvalidValue;
  int wgs_per_row = 0;
  if (sk_partials) {
    const unsigned long long per_row = 1ull << d.j_bits;  // instances/row
    if (per_row % F != 0) return hipErrorInvalidValue;
    wgs_per_row = (int)(per_row / F);
  }
  if (sign < 0) {
    if (sk_partials)
      hipLaunchKernelGGL((k_fft_dif_final<-1, true>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, tw_n, sk_partials,
                         wgs_per_row);
    else
      hipLaunchKernelGGL((k_fft_dif_final<-1, false>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, tw_n, nullptr, 0);
  } else {
    if (sk_partials)
      hipLaunchKernelGGL((k_fft_dif_final<1, true>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, tw_n, sk_partials,
                         wgs_per_row);
    else
      hipLaunchKernelGGL((k_fft_dif_final<1, false>), dim3(grid), dim3(256),
                         lds_bytes, stream, in, out, d, tw_n, nullptr, 0);
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t sk_combine_partials(const float2* partials, size_t rows,
                               int wgs_per_row, float2* s2s4,
                               hipStream_t stream) {
  hipLaunchKernelGGL(k_sk_combine_partials, grid_for(rows), dim3(kBlock), 0,
                     stream, partials, rows, wgs_per_row, s2s4);
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
