// Unpack kernels: packed baseband bytes → float32 samples, FFT window fused.
//
// Capability parity with reference userspace/include/srtb/unpack.hpp:43-403
// (1/2/4-bit MSB-first unsigned fields; 8/16/32-bit signed/unsigned casts;
// 2-pol per-sample interleave; SNAP-1 "1 1 2 2"; GZNU A1 4-byte-word
// deinterleave with offset-binary fix), redesigned for CDNA4:
//  - sub-byte formats: one byte per lane (wave-contiguous float4/float2
//    stores; see the k_unpack_subbyte note); byte casts: one uint32 per
//    lane with a float4 store
//  - wave64-sized blocks, grid-stride loops capped per G11.

#include "common.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

namespace {

template <bool kWindow>
__device__ inline float wmul(const float* __restrict__ w, size_t i, float v) {
  if constexpr (kWindow) return v * w[i];
  return v;
}

// ---- sub-byte unpack: one BYTE per lane per iteration ----
// bits per sample B in {1,2,4}: byte yields 8/B samples, MSB-first.
//
// Store orientation matters more than load width here: a word-per-lane
// variant emitted 4 sequential float4 stores per lane (16 B per lane at
// 64 B lane stride per instruction), which triggers read-for-ownership on
// every partially-written line — measured 2.7 TB/s vs 4.9 for the 8-bit
// cast.  Byte-per-lane makes every store instruction wave-contiguous
// (64 lanes × consecutive float4/float2), and the narrow byte loads only
// re-read the 0.25 GB input.
template <int NBITS, bool kWindow>
__global__ void k_unpack_subbyte(const uint8_t* __restrict__ in,
                                 float* __restrict__ out, size_t n_bytes,
                                 const float* __restrict__ window) {
  constexpr int per_byte = 8 / NBITS;
  constexpr uint32_t mask = (1u << NBITS) - 1u;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
#pragma unroll 4
  for (size_t b = (size_t)blockIdx.x * blockDim.x + threadIdx.x; b < n_bytes;
       b += stride) {
    const uint32_t bv = in[b];
    const size_t base = b * per_byte;
    float vals[per_byte];
#pragma unroll
    for (int i = 0; i < per_byte; ++i) {
      const uint32_t field = (bv >> ((per_byte - 1 - i) * NBITS)) & mask;
      vals[i] = wmul<kWindow>(window, base + i, (float)field);
    }
    if constexpr (NBITS == 2) {
      reinterpret_cast<float4*>(out + base)[0] =
          make_float4(vals[0], vals[1], vals[2], vals[3]);
    } else if constexpr (NBITS == 4) {
      reinterpret_cast<float2*>(out + base)[0] =
          make_float2(vals[0], vals[1]);
    } else {  // NBITS == 1: two float4 stores per byte
      float4* o4 = reinterpret_cast<float4*>(out + base);
      o4[0] = make_float4(vals[0], vals[1], vals[2], vals[3]);
      o4[1] = make_float4(vals[4], vals[5], vals[6], vals[7]);
    }
  }
}

// ---- sub-byte unpack, wave-block variant: dense dword loads + __shfl ----
// Byte-per-lane above fixed the STORE side but still loads one byte per
// lane (64 B touched per 64-lane load instruction).  Here each wave owns a
// 256-byte block per iteration: every lane loads one uint32 (dense 256 B
// run), then __shfl redistributes so each lane decodes the byte whose
// OUTPUT vector it stores — both loads and stores are wave-contiguous.
template <int NBITS, bool kWindow>
__global__ void k_unpack_subbyte_w(const uint32_t* __restrict__ in,
                                   float* __restrict__ out, size_t n_blocks,
                                   const float* __restrict__ window) {
  constexpr int per_byte = 8 / NBITS;
  constexpr uint32_t mask = (1u << NBITS) - 1u;
  const int lane = threadIdx.x & 63;
  const size_t wave = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const size_t wstride = ((size_t)gridDim.x * blockDim.x) >> 6;
  // unroll keeps ≥2 block loads in flight (one dword feeds a whole block of
  // stores, so without it the loop is a single-load latency chain)
#pragma unroll 2
  for (size_t blk = wave; blk < n_blocks; blk += wstride) {
    const uint32_t myw = in[blk * 64 + lane];
    const size_t sample_base = blk * (256 * per_byte);
    if constexpr (NBITS == 1) {
      // float4-centric: float4 fi covers half a byte; 512 float4s per block
#pragma unroll
      for (int r = 0; r < 8; ++r) {
        const int fi = 64 * r + lane;
        const uint32_t dw = __shfl(myw, 8 * r + (lane >> 3), 64);
        const uint32_t bv = (dw >> (8 * ((lane >> 1) & 3))) & 0xffu;
        const int half = lane & 1;  // low or high 4 samples of the byte
        const size_t sbase = sample_base + (size_t)fi * 4;
        float4 o;
        float* op = &o.x;
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          const int s = half * 4 + i;  // MSB-first within the byte
          op[i] = wmul<kWindow>(window, sbase + i,
                                (float)((bv >> (7 - s)) & 1u));
        }
        reinterpret_cast<float4*>(out + sbase)[0] = o;
      }
    } else if constexpr (NBITS == 2) {
      // byte-centric: one float4 per byte, 256 bytes per block, 4 rounds
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int bi = 64 * r + lane;
        const uint32_t dw = __shfl(myw, 16 * r + (lane >> 2), 64);
        const uint32_t bv = (dw >> (8 * (lane & 3))) & 0xffu;
        const size_t sbase = sample_base + (size_t)bi * per_byte;
        float4 o;
        float* op = &o.x;
#pragma unroll
        for (int i = 0; i < 4; ++i)
          op[i] = wmul<kWindow>(window, sbase + i,
                                (float)((bv >> ((3 - i) * 2)) & mask));
        reinterpret_cast<float4*>(out + sbase)[0] = o;
      }
    } else {
      // NBITS == 4, float4-centric: one float4 covers two bytes, 2 rounds
#pragma unroll
      for (int r = 0; r < 2; ++r) {
        const int fi = 64 * r + lane;
        const uint32_t dw = __shfl(myw, 32 * r + (lane >> 1), 64);
        const uint32_t b0 = (dw >> (8 * ((lane & 1) * 2))) & 0xffu;
        const uint32_t b1 = (dw >> (8 * ((lane & 1) * 2 + 1))) & 0xffu;
        const size_t sbase = sample_base + (size_t)fi * 4;
        float4 o;
        o.x = wmul<kWindow>(window, sbase + 0, (float)((b0 >> 4) & mask));
        o.y = wmul<kWindow>(window, sbase + 1, (float)(b0 & mask));
        o.z = wmul<kWindow>(window, sbase + 2, (float)((b1 >> 4) & mask));
        o.w = wmul<kWindow>(window, sbase + 3, (float)(b1 & mask));
        reinterpret_cast<float4*>(out + sbase)[0] = o;
      }
    }
  }
}

// ---- byte casts: one uint32 = 4 samples per lane ----
template <bool kSigned, bool kWindow>
__global__ void k_unpack_cast8(const uint32_t* __restrict__ in,
                               float* __restrict__ out, size_t n_words,
                               const float* __restrict__ window) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t w = (size_t)blockIdx.x * blockDim.x + threadIdx.x; w < n_words;
       w += stride) {
    const uint32_t v = in[w];
    const size_t base = w * 4;
    float4 o;
    float* op = &o.x;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const uint32_t b = (v >> (8 * i)) & 0xffu;
      const float f = kSigned ? (float)(int8_t)b : (float)b;
      op[i] = wmul<kWindow>(window, base + i, f);
    }
    reinterpret_cast<float4*>(out + base)[0] = o;
  }
}

template <typename T, bool kWindow>
__global__ void k_unpack_cast(const T* __restrict__ in, float* __restrict__ out,
                              size_t n, const float* __restrict__ window) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = wmul<kWindow>(window, i, (float)in[i]);
}

// ---- 2-pol per-sample interleave: [p0 p1 p0 p1 ...] int8 ----
// one lane consumes 8 bytes (uint2) = 4 samples per pol, float4 stores.
template <bool kWindow>
__global__ void k_unpack_2pol(const uint2* __restrict__ in,
                              float* __restrict__ out0,
                              float* __restrict__ out1, size_t n_groups,
                              const float* __restrict__ window) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t g = (size_t)blockIdx.x * blockDim.x + threadIdx.x; g < n_groups;
       g += stride) {
    const uint2 v = in[g];
    const size_t base = g * 4;
    float4 a, b;
    float* ap = &a.x;
    float* bp = &b.x;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      const uint32_t word = (i == 0) ? v.x : v.y;
      ap[2 * i + 0] = (float)(int8_t)(word & 0xff);
      bp[2 * i + 0] = (float)(int8_t)((word >> 8) & 0xff);
      ap[2 * i + 1] = (float)(int8_t)((word >> 16) & 0xff);
      bp[2 * i + 1] = (float)(int8_t)((word >> 24) & 0xff);
    }
    if constexpr (kWindow) {
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        ap[i] *= window[base + i];
        bp[i] *= window[base + i];
      }
    }
    reinterpret_cast<float4*>(out0 + base)[0] = a;
    reinterpret_cast<float4*>(out1 + base)[0] = b;
  }
}

// ---- SNAP-1 "1 1 2 2": uint32 = [s0p0 s1p0 s0p1 s1p1] ----
// one lane consumes 8 bytes = 4 samples per pol.
template <bool kWindow>
__global__ void k_unpack_snap1(const uint2* __restrict__ in,
                               float* __restrict__ out0,
                               float* __restrict__ out1, size_t n_groups,
                               const float* __restrict__ window) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t g = (size_t)blockIdx.x * blockDim.x + threadIdx.x; g < n_groups;
       g += stride) {
    const uint2 v = in[g];
    const size_t base = g * 4;
    float4 a, b;
    a.x = (float)(int8_t)(v.x & 0xff);
    a.y = (float)(int8_t)((v.x >> 8) & 0xff);
    b.x = (float)(int8_t)((v.x >> 16) & 0xff);
    b.y = (float)(int8_t)((v.x >> 24) & 0xff);
    a.z = (float)(int8_t)(v.y & 0xff);
    a.w = (float)(int8_t)((v.y >> 8) & 0xff);
    b.z = (float)(int8_t)((v.y >> 16) & 0xff);
    b.w = (float)(int8_t)((v.y >> 24) & 0xff);
    if constexpr (kWindow) {
      float* ap = &a.x;
      float* bp = &b.x;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        ap[i] *= window[base + i];
        bp[i] *= window[base + i];
      }
    }
    reinterpret_cast<float4*>(out0 + base)[0] = a;
    reinterpret_cast<float4*>(out1 + base)[0] = b;
  }
}

// ---- GZNU A1: 4-byte words cycling over NS streams ----
// one lane consumes NS consecutive words (uint32 each) = 4 samples per stream.
template <int NS, bool kXor80, bool kWindow>
__global__ void k_unpack_gznupsr(const uint32_t* __restrict__ in,
                                 float* __restrict__ o0, float* __restrict__ o1,
                                 float* __restrict__ o2, float* __restrict__ o3,
                                 size_t n_groups,
                                 const float* __restrict__ window) {
  float* outs[4] = {o0, o1, o2, o3};
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t g = (size_t)blockIdx.x * blockDim.x + threadIdx.x; g < n_groups;
       g += stride) {
    const size_t base = g * 4;
#pragma unroll
    for (int s = 0; s < NS; ++s) {
      uint32_t w = in[g * NS + s];
      if constexpr (kXor80) w ^= 0x80808080u;
      float4 o;
      o.x = (float)(int8_t)(w & 0xff);
      o.y = (float)(int8_t)((w >> 8) & 0xff);
      o.z = (float)(int8_t)((w >> 16) & 0xff);
      o.w = (float)(int8_t)((w >> 24) & 0xff);
      if constexpr (kWindow) {
        o.x *= window[base];
        o.y *= window[base + 1];
        o.z *= window[base + 2];
        o.w *= window[base + 3];
      }
      if (outs[s]) reinterpret_cast<float4*>(outs[s] + base)[0] = o;
    }
  }
}

// cosine-sum window coefficient table: coef[i] = a0 - a1*cos(2*pi*i/(n-1))
// (reference fft/fft_window.hpp:27-83; hann a0=a1=0.5, hamming 25/46, 21/46)
__global__ void k_build_window(float* __restrict__ coef, size_t n, double a0,
                               double a1) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  const double inv = (n > 1) ? 1.0 / (double)(n - 1) : 0.0;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    coef[i] = (float)(a0 - a1 * cos(2.0 * M_PI * (double)i * inv));
}

}  // namespace

hipError_t build_window(float* coef, size_t n, int kind, hipStream_t stream) {
  double a0 = 1.0, a1 = 0.0;
  if (kind == 1) { a0 = 0.5; a1 = 0.5; }                    // hann
  else if (kind == 2) { a0 = 25.0 / 46.0; a1 = 21.0 / 46.0; }  // hamming
  else if (kind != 0) return hipErrorInvalidValue;
  hipLaunchKernelGGL(k_build_window, grid_for(n), dim3(kBlock), 0, stream,
                     coef, n, a0, a1);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t unpack(const uint8_t* in, float* out, size_t out_count, int nbits,
                  const float* window, hipStream_t stream) {
  const bool w = window != nullptr;
  switch (nbits) {
    case 1:
    case 2:
    case 4: {
      const size_t n_bytes = out_count * nbits / 8;
      // main body: whole 256-byte wave blocks (dense loads); byte-per-lane
      // tail kernel covers the remainder (only for non-multiple-of-256
      // inputs — never on power-of-two block sizes)
      const size_t n_blocks = n_bytes >> 8;
      const size_t tail_off = n_blocks << 8;
      const size_t tail = n_bytes - tail_off;
#define CASE(B)                                                              \
  if (nbits == B) {                                                          \
    constexpr int PB = 8 / B;                                                \
    if (n_blocks) {                                                          \
      const dim3 g = grid_for(n_blocks * 64);                                \
      if (w)                                                                 \
        hipLaunchKernelGGL((k_unpack_subbyte_w<B, true>), g, dim3(kBlock),   \
                           0, stream,                                        \
                           reinterpret_cast<const uint32_t*>(in), out,       \
                           n_blocks, window);                                \
      else                                                                   \
        hipLaunchKernelGGL((k_unpack_subbyte_w<B, false>), g, dim3(kBlock),  \
                           0, stream,                                        \
                           reinterpret_cast<const uint32_t*>(in), out,       \
                           n_blocks, window);                                \
    }                                                                        \
    if (tail) {                                                              \
      const dim3 gt = grid_for(tail);                                        \
      const float* wt = w ? window + tail_off * PB : nullptr;                \
      if (w)                                                                 \
        hipLaunchKernelGGL((k_unpack_subbyte<B, true>), gt, dim3(kBlock), 0, \
                           stream, in + tail_off, out + tail_off * PB, tail, \
                           wt);                                              \
      else                                                                   \
        hipLaunchKernelGGL((k_unpack_subbyte<B, false>), gt, dim3(kBlock),   \
                           0, stream, in + tail_off, out + tail_off * PB,    \
                           tail, wt);                                        \
    }                                                                        \
  }
      CASE(1) CASE(2) CASE(4)
#undef CASE
      break;
    }
    case 8:
    case -8: {
      const size_t n_words = out_count / 4;
      const dim3 g = grid_for(n_words);
      auto* in32 = reinterpret_cast<const uint32_t*>(in);
      if (nbits < 0) {
        if (w)
          hipLaunchKernelGGL((k_unpack_cast8<true, true>), g, dim3(kBlock), 0,
                             stream, in32, out, n_words, window);
        else
          hipLaunchKernelGGL((k_unpack_cast8<true, false>), g, dim3(kBlock), 0,
                             stream, in32, out, n_words, window);
      } else {
        if (w)
          hipLaunchKernelGGL((k_unpack_cast8<false, true>), g, dim3(kBlock), 0,
                             stream, in32, out, n_words, window);
        else
          hipLaunchKernelGGL((k_unpack_cast8<false, false>), g, dim3(kBlock),
                             0, stream, in32, out, n_words, window);
      }
      break;
    }
#define SRTB_CAST_CASE(BITS, T)                                              \
    case BITS:                                                               \
      if (w)                                                                 \
        hipLaunchKernelGGL((k_unpack_cast<T, true>), grid_for(out_count),    \
                           dim3(kBlock), 0, stream,                          \
                           reinterpret_cast<const T*>(in), out, out_count,   \
                           window);                                          \
      else                                                                   \
        hipLaunchKernelGGL((k_unpack_cast<T, false>), grid_for(out_count),   \
                           dim3(kBlock), 0, stream,                          \
                           reinterpret_cast<const T*>(in), out, out_count,   \
                           window);                                          \
      break;
    SRTB_CAST_CASE(16, uint16_t)
    SRTB_CAST_CASE(-16, int16_t)
    SRTB_CAST_CASE(32, uint32_t)
    SRTB_CAST_CASE(-32, int32_t)
#undef SRTB_CAST_CASE
    default:
      return hipErrorInvalidValue;
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t unpack_interleaved_2pol(const int8_t* in, float* out0, float* out1,
                                   size_t count_per_pol, const float* window,
                                   hipStream_t stream) {
  const size_t n_groups = count_per_pol / 4;
  const dim3 g = grid_for(n_groups);
  auto* in2 = reinterpret_cast<const uint2*>(in);
  if (window)
    hipLaunchKernelGGL((k_unpack_2pol<true>), g, dim3(kBlock), 0, stream, in2,
                       out0, out1, n_groups, window);
  else
    hipLaunchKernelGGL((k_unpack_2pol<false>), g, dim3(kBlock), 0, stream, in2,
                       out0, out1, n_groups, window);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t unpack_naocpsr_snap1(const int8_t* in, float* out0, float* out1,
                                size_t count_per_pol, const float* window,
                                hipStream_t stream) {
  const size_t n_groups = count_per_pol / 4;
  const dim3 g = grid_for(n_groups);
  auto* in2 = reinterpret_cast<const uint2*>(in);
  if (window)
    hipLaunchKernelGGL((k_unpack_snap1<true>), g, dim3(kBlock), 0, stream, in2,
                       out0, out1, n_groups, window);
  else
    hipLaunchKernelGGL((k_unpack_snap1<false>), g, dim3(kBlock), 0, stream,
                       in2, out0, out1, n_groups, window);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t unpack_gznupsr_a1(const uint8_t* in, float* out0, float* out1,
                             float* out2, float* out3, int n_streams,
                             size_t count_per_stream, const float* window,
                             hipStream_t stream) {
  const size_t n_groups = count_per_stream / 4;
  const dim3 g = grid_for(n_groups);
  auto* in32 = reinterpret_cast<const uint32_t*>(in);
  const bool w = window != nullptr;
  if (n_streams == 2) {
    if (w)
      hipLaunchKernelGGL((k_unpack_gznupsr<2, false, true>), g, dim3(kBlock),
                         0, stream, in32, out0, out1, out2, out3, n_groups,
                         window);
    else
      hipLaunchKernelGGL((k_unpack_gznupsr<2, false, false>), g, dim3(kBlock),
                         0, stream, in32, out0, out1, out2, out3, n_groups,
                         window);
  } else if (n_streams == 4) {
    if (w)
      hipLaunchKernelGGL((k_unpack_gznupsr<4, true, true>), g, dim3(kBlock), 0,
                         stream, in32, out0, out1, out2, out3, n_groups,
                         window);
    else
      hipLaunchKernelGGL((k_unpack_gznupsr<4, true, false>), g, dim3(kBlock),
                         0, stream, in32, out0, out1, out2, out3, n_groups,
                         window);
  } else {
    return hipErrorInvalidValue;
  }
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

}  // namespace srtb_hip
