// hipFFT plan management for the srtb_amd engine.
//
// Capability parity with the reference's fft_1d_dispatcher / shared-work-area
// design (fft/fft.hpp:56-160, fft/cufft_like_wrapper.hpp:42-161): plans are
// created once per (type, n, batch), auto-allocation is disabled and every
// plan of an engine instance shares one scratch arena sized to the maximum
// requirement.  A plan binds to the engine slot's HIP stream.
#pragma once

#include <hip/hip_runtime.h>
#include <hipfft/hipfft.h>

#include <cstddef>
#include <stdexcept>
#include <string>

namespace srtb_hip {

inline void check_fft(hipfftResult r, const char* what) {
  if (r != HIPFFT_SUCCESS)
    throw std::runtime_error(std::string("hipFFT error ") +
                             std::to_string((int)r) + " in " + what);
}

inline void check_hip(hipError_t e, const char* what) {
  if (e != hipSuccess)
    throw std::runtime_error(std::string("HIP error: ") +
                             hipGetErrorString(e) + " in " + what);
}

// One R2C plan (n real → n/2+1 complex) and one batched C2C plan
// (len × batch), sharing a work area.  Owned by one engine slot.
class FftPlanSet {
 public:
  FftPlanSet() = default;
  FftPlanSet(const FftPlanSet&) = delete;
  FftPlanSet& operator=(const FftPlanSet&) = delete;

  void create(size_t n_real, size_t c2c_len, size_t c2c_batch,
              hipStream_t stream) {
    destroy();
    size_t ws_r2c = 0, ws_c2c = 0;
    long long n1[1] = {(long long)n_real};
    check_fft(hipfftCreate(&r2c_), "hipfftCreate r2c");
    check_fft(hipfftSetAutoAllocation(r2c_, 0), "SetAutoAllocation r2c");
    check_fft(hipfftMakePlanMany64(r2c_, 1, n1, nullptr, 1, 0, nullptr, 1, 0,
                                   HIPFFT_R2C, 1, &ws_r2c),
              "MakePlanMany64 r2c");

    long long n2[1] = {(long long)c2c_len};
    check_fft(hipfftCreate(&c2c_), "hipfftCreate c2c");
    check_fft(hipfftSetAutoAllocation(c2c_, 0), "SetAutoAllocation c2c");
    check_fft(hipfftMakePlanMany64(c2c_, 1, n2, nullptr, 1,
                                   (long long)c2c_len, nullptr, 1,
                                   (long long)c2c_len, HIPFFT_C2C,
                                   (long long)c2c_batch, &ws_c2c),
              "MakePlanMany64 c2c");

    work_size_ = ws_r2c > ws_c2c ? ws_r2c : ws_c2c;
    if (work_size_) {
      check_hip(hipMalloc(&work_area_, work_size_), "hipMalloc fft work");
      check_fft(hipfftSetWorkArea(r2c_, work_area_), "SetWorkArea r2c");
      check_fft(hipfftSetWorkArea(c2c_, work_area_), "SetWorkArea c2c");
    }
    check_fft(hipfftSetStream(r2c_, stream), "SetStream r2c");
    check_fft(hipfftSetStream(c2c_, stream), "SetStream c2c");
  }

  // C2C plan only (the mixed "auto" backend: native forward FFT + hipFFT
  // batched backward) — skips the R2C plan, whose work area for a 2^30-point
  // transform would waste gigabytes on a plan that never runs.
  void create_c2c_only(size_t c2c_len, size_t c2c_batch, hipStream_t stream) {
    destroy();
    size_t ws_c2c = 0;
    long long n2[1] = {(long long)c2c_len};
    check_fft(hipfftCreate(&c2c_), "hipfftCreate c2c");
    check_fft(hipfftSetAutoAllocation(c2c_, 0), "SetAutoAllocation c2c");
    check_fft(hipfftMakePlanMany64(c2c_, 1, n2, nullptr, 1,
                                   (long long)c2c_len, nullptr, 1,
                                   (long long)c2c_len, HIPFFT_C2C,
                                   (long long)c2c_batch, &ws_c2c),
              "MakePlanMany64 c2c");
    work_size_ = ws_c2c;
    if (work_size_) {
      check_hip(hipMalloc(&work_area_, work_size_), "hipMalloc fft work");
      check_fft(hipfftSetWorkArea(c2c_, work_area_), "SetWorkArea c2c");
    }
    check_fft(hipfftSetStream(c2c_, stream), "SetStream c2c");
  }

  void exec_r2c(float* in, float2* out) {
    check_fft(hipfftExecR2C(r2c_, in, reinterpret_cast<hipfftComplex*>(out)),
              "ExecR2C");
  }

  void exec_c2c_backward(float2* inout) {
    check_fft(hipfftExecC2C(c2c_, reinterpret_cast<hipfftComplex*>(inout),
                            reinterpret_cast<hipfftComplex*>(inout),
                            HIPFFT_BACKWARD),
              "ExecC2C backward");
  }

  size_t work_size() const { return work_size_; }

  void destroy() {
    if (r2c_) (void)hipfftDestroy(r2c_), r2c_ = 0;
    if (c2c_) (void)hipfftDestroy(c2c_), c2c_ = 0;
    if (work_area_) (void)hipFree(work_area_), work_area_ = nullptr;
    work_size_ = 0;
  }

  ~FftPlanSet() { destroy(); }

 private:
  hipfftHandle r2c_ = 0, c2c_ = 0;
  void* work_area_ = nullptr;
  size_t work_size_ = 0;
};

}  // namespace srtb_hip
