// Shared device helpers for srtb_amd HIP kernels (gfx950, wave64).
#pragma once

#include <hip/hip_runtime.h>
#include <cstddef>
#include <cstdint>

namespace srtb_hip {

constexpr int kBlock = 256;          // multiple of wave64
constexpr int kMaxBlocks = 4096;     // grid-stride cap (G11: ~256 CU * 8-16)

inline dim3 grid_for(size_t n, int block = kBlock, int cap = kMaxBlocks) {
  size_t b = (n + block - 1) / block;
  if (b > static_cast<size_t>(cap)) b = cap;
  if (b == 0) b = 1;
  return dim3(static_cast<uint32_t>(b));
}

#define SRTB_CHECK_LAUNCH()                        \
  do {                                             \
    hipError_t err_ = hipGetLastError();           \
    if (err_ != hipSuccess) return err_;           \
  } while (0)

// ---- wave64 + block reductions (fp32 lanes, fp64 block output) ----

__device__ inline float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

__device__ inline double wave_reduce_sum(double v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

// Block-level sum over kBlock threads; returns valid value on thread 0.
template <typename T>
__device__ inline T block_reduce_sum(T v) {
  __shared__ T lds[kBlock / 64];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wave] = v;
  __syncthreads();
  T out = T(0);
  if (wave == 0) {
    out = (lane < kBlock / 64) ? lds[lane] : T(0);
    out = wave_reduce_sum(out);
  }
  __syncthreads();
  return out;
}

__device__ inline float norm2(float2 c) { return c.x * c.x + c.y * c.y; }

// Coherent-dedispersion phase factor (reference phase_factor_v3,
// coherent_dedispersion.hpp:133-150): k = D*1e6*dm/f*((f-f_c)/f_c)^2 with
// |k| up to ~1e9, so everything before the wrap stays in fp64.
constexpr double kDispersionConstantMHz = 4.148808e3;

__device__ inline float2 srtb_dedisp_factor(size_t i, double f_min,
                                            double f_c, double df,
                                            double dm) {
  const double f = f_min + df * (double)i;
  const double r = (f - f_c) / f_c;
  const double k = (kDispersionConstantMHz * 1e6) * dm / f * (r * r);
  double k_int;
  const double k_frac = modf(k, &k_int);
  // |k| reaches ~1e9 so the reduction above must be fp64, but the wrapped
  // phase is in (-2pi, 2pi): fast f32 sincos is accurate to ~1.5e-6 rad
  // here, comparable to the fp32 storage precision, at a fraction of the
  // fp64 sincos VALU cost.
  const float phi = (float)(-2.0 * M_PI * k_frac);
  float s, c;
  __sincosf(phi, &s, &c);
  return make_float2(c, s);
}

}  // namespace srtb_hip
