// Display-path and misc kernels: waterfall resample, normalization, pixmap,
// running mean, correlator pointwise product.
//
// Reference semantics: spectrum/simplify_spectrum.hpp:276-731 (resample v2/v3
// math: fractional-coverage area average; one workgroup per output pixel at
// wg=64 — measured fastest on wave64 hardware in the reference, a natural fit
// for one CDNA4 wavefront, so the tree reduce is pure __shfl, no LDS),
// algorithm/running_mean.hpp:31-77, src/correlator.cpp:116-140.

#include "common.h"
#include "../include/srtb_kernels.h"

namespace srtb_hip {

namespace {

// one wave (64 lanes) per output pixel; lanes stride the x (time) range,
// y (row) loop serial with fractional edge weights.
__global__ void __launch_bounds__(64)
    k_resample_power(const float2* __restrict__ wf, size_t rows, size_t len,
                     float* __restrict__ out, int out_h, int out_w) {
  const int pix = blockIdx.x;
  const int oy = pix / out_w;
  const int ox = pix - oy * out_w;
  const double ys = (double)rows / out_h, xs = (double)len / out_w;
  const double y0 = oy * ys, y1 = (oy + 1) * ys;
  const double x0 = ox * xs, x1 = (ox + 1) * xs;
  const long iy0 = (long)floor(y0), iy1 = min((long)ceil(y1), (long)rows);
  const long ix0 = (long)floor(x0), ix1 = min((long)ceil(x1), (long)len);
  float acc = 0.0f;
  for (long y = iy0; y < iy1; ++y) {
    const float wy =
        (float)(fmin((double)y + 1.0, y1) - fmax((double)y, y0));
    const float2* rp = wf + (size_t)y * len;
    for (long x = ix0 + (long)threadIdx.x; x < ix1; x += 64) {
      const float wx =
          (float)(fmin((double)x + 1.0, x1) - fmax((double)x, x0));
      acc += wy * wx * norm2(rp[x]);
    }
  }
  acc = wave_reduce_sum(acc);
  if (threadIdx.x == 0)
    out[pix] = acc / (float)((y1 - y0) * (x1 - x0));
}

__global__ void k_normalize_by_mean(float* __restrict__ img, size_t n,
                                    const double* __restrict__ sum) {
  const double mean = *sum / (double)n;
  const float scale = (mean != 0.0) ? (float)(1.0 / (2.0 * mean)) : 1.0f;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    img[i] *= scale;
}

__global__ void k_pixmap(const float* __restrict__ in,
                         uint32_t* __restrict__ out, size_t n, uint32_t c0,
                         uint32_t c1, uint32_t cover) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float x = in[i];
    if (x < 0.0f || x > 1.0f || !isfinite(x)) {
      out[i] = cover;
      continue;
    }
    uint32_t px = 0;
#pragma unroll
    for (int sh = 0; sh < 32; sh += 8) {
      const float a = (float)((c0 >> sh) & 0xff);
      const float b = (float)((c1 >> sh) & 0xff);
      const int v = (int)roundf(a + (b - a) * x);
      px |= ((uint32_t)min(max(v, 0), 255)) << sh;
    }
    out[i] = px;
  }
}

__global__ void k_running_mean_init(const float* __restrict__ data,
                                    size_t nchan, size_t windowsize,
                                    float* __restrict__ ave) {
  const size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= nchan) return;
  float s = 0.0f;
  for (size_t k = 0; k < windowsize; ++k) s += data[k * nchan + j];
  ave[j] = s / (float)windowsize;
}

__global__ void k_running_mean(const float* __restrict__ data, size_t nsamp,
                               size_t nchan, uint8_t* __restrict__ out,
                               size_t windowsize, float* __restrict__ ave) {
  const size_t j = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= nchan) return;
  float a = ave[j];
  for (size_t i = windowsize; i < nsamp; ++i) {
    const float head = data[(i - windowsize) * nchan + j];
    const float tail = data[i * nchan + j];
    out[(i - windowsize) * nchan + j] = head > a ? 1 : 0;
    a += (tail - head) / (float)windowsize;
  }
  for (size_t i = 0; i < windowsize; ++i) {
    const float head = data[(nsamp + i - windowsize) * nchan + j];
    const float tail = data[(nsamp - i - 1) * nchan + j];
    out[(i + nsamp - windowsize) * nchan + j] = head > a ? 1 : 0;
    a += (tail - head) / (float)windowsize;
  }
  ave[j] = a;
}

__global__ void k_complex_abs(const float2* __restrict__ x,
                              float* __restrict__ mag, size_t n) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    mag[i] = sqrtf(norm2(x[i]));
}

__global__ void k_correlate(const float2* __restrict__ f1,
                            const float2* __restrict__ f2,
                            float2* __restrict__ corr, float* __restrict__ mag,
                            size_t n, float scale) {
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float2 a = f1[i], b = f2[i];
    // a * conj(b) * scale
    const float2 c = make_float2(scale * (a.x * b.x + a.y * b.y),
                                 scale * (a.y * b.x - a.x * b.y));
    corr[i] = c;
    if (mag) mag[i] = sqrtf(c.x * c.x + c.y * c.y);
  }
}

}  // namespace

hipError_t resample_power_2d(const float2* wf, size_t rows, size_t len,
                             float* out, int out_h, int out_w,
                             hipStream_t stream) {
  const size_t pixels = (size_t)out_h * out_w;
  hipLaunchKernelGGL(k_resample_power, dim3((uint32_t)pixels), dim3(64), 0,
                     stream, wf, rows, len, out, out_h, out_w);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t normalize_by_mean(float* img, size_t n, const double* sum,
                             hipStream_t stream) {
  hipLaunchKernelGGL(k_normalize_by_mean, grid_for(n), dim3(kBlock), 0, stream,
                     img, n, sum);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t generate_pixmap(const float* intensity, uint32_t* out, size_t n,
                           uint32_t color0, uint32_t color1,
                           uint32_t color_overflow, hipStream_t stream) {
  hipLaunchKernelGGL(k_pixmap, grid_for(n), dim3(kBlock), 0, stream, intensity,
                     out, n, color0, color1, color_overflow);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t running_mean_init(const float* data, size_t nsamp, size_t nchan,
                             size_t windowsize, float* ave,
                             hipStream_t stream) {
  (void)nsamp;
  hipLaunchKernelGGL(k_running_mean_init, grid_for(nchan), dim3(kBlock), 0,
                     stream, data, nchan, windowsize, ave);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t running_mean(const float* data, size_t nsamp, size_t nchan,
                        uint8_t* out, size_t windowsize, float* ave,
                        hipStream_t stream) {
  hipLaunchKernelGGL(k_running_mean, grid_for(nchan), dim3(kBlock), 0, stream,
                     data, nsamp, nchan, out, windowsize, ave);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t complex_abs(const float2* x, float* mag, size_t n,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_complex_abs, grid_for(n), dim3(kBlock), 0, stream, x,
                     mag, n);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

hipError_t correlate_pointwise(const float2* f1, const float2* f2,
                               float2* corr, float* mag, size_t n, float scale,
                               hipStream_t stream) {
  hipLaunchKernelGGL(k_correlate, grid_for(n), dim3(kBlock), 0, stream, f1, f2,
                     corr, mag, n, scale);
  SRTB_CHECK_LAUNCH();
  return hipSuccess;
}

}  // namespace srtb_hip
