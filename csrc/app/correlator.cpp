// srtb-correlator — offline cross-correlation of two recordings
// (reference src/correlator.cpp:35-152): unpack → R2C FFT → scale·F1·conj(F2)
// → backward C2C → |corr| → float32 output file.

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstring>
#include <fstream>
#include <string>
#include <vector>

#include "../fft/fft_plans.h"
#include "../include/srtb_kernels.h"
#include "config.h"
#include "runtime.h"

using namespace srtb_app;

int main(int argc, char** argv) {
  install_termination_handler();
  if (argc < 4) {
    std::fprintf(stderr,
                 "usage: srtb-correlator <file1> <file2> <out.bin> "
                 "[--nbits -8] [--count '2 ** 20'] [--offset-bytes N]\n");
    return 2;
  }
  std::string f1 = argv[1], f2 = argv[2], out_path = argv[3];
  int nbits = -8;
  size_t count = 1 << 20, offset = 0;
  for (int i = 4; i + 1 < argc; i += 2) {
    const std::string a = argv[i];
    if (a == "--nbits") nbits = (int)Expr::eval_int(argv[i + 1]);
    else if (a == "--count") count = (size_t)Expr::eval_int(argv[i + 1]);
    else if (a == "--offset-bytes") offset = (size_t)Expr::eval_int(argv[i + 1]);
  }
  const size_t in_bytes = count * (size_t)std::abs(nbits) / 8;

  std::vector<uint8_t> h1(in_bytes), h2(in_bytes);
  for (auto* pair : {&h1, &h2}) {
    const std::string& path = (pair == &h1) ? f1 : f2;
    std::ifstream f(path, std::ios::binary);
    if (!f) { std::fprintf(stderr, "cannot open %s\n", path.c_str()); return 1; }
    f.seekg((std::streamoff)offset);
    f.read(reinterpret_cast<char*>(pair->data()), (std::streamsize)in_bytes);
    if ((size_t)f.gcount() < in_bytes) {
      std::fprintf(stderr, "%s too short\n", path.c_str());
      return 1;
    }
  }

  const size_t nc = count / 2;
  hipStream_t st;
  srtb_hip::check_hip(hipStreamCreate(&st), "stream");
  uint8_t *d_r1, *d_r2;
  float *d_s1, *d_s2, *d_mag;
  float2 *d_f1, *d_f2;
  srtb_hip::check_hip(hipMalloc(&d_r1, in_bytes), "a");
  srtb_hip::check_hip(hipMalloc(&d_r2, in_bytes), "a");
  srtb_hip::check_hip(hipMalloc(&d_s1, count * sizeof(float)), "a");
  srtb_hip::check_hip(hipMalloc(&d_s2, count * sizeof(float)), "a");
  srtb_hip::check_hip(hipMalloc(&d_f1, (nc + 1) * sizeof(float2)), "a");
  srtb_hip::check_hip(hipMalloc(&d_f2, (nc + 1) * sizeof(float2)), "a");
  srtb_hip::check_hip(hipMalloc(&d_mag, nc * sizeof(float)), "a");
  srtb_hip::check_hip(
      hipMemcpyAsync(d_r1, h1.data(), in_bytes, hipMemcpyHostToDevice, st),
      "h2d");
  srtb_hip::check_hip(
      hipMemcpyAsync(d_r2, h2.data(), in_bytes, hipMemcpyHostToDevice, st),
      "h2d");
  srtb_hip::check_hip(srtb_hip::unpack(d_r1, d_s1, count, nbits, nullptr, st),
                      "unpack");
  srtb_hip::check_hip(srtb_hip::unpack(d_r2, d_s2, count, nbits, nullptr, st),
                      "unpack");

  srtb_hip::FftPlanSet plans;
  plans.create(count, nc, 1, st);
  plans.exec_r2c(d_s1, d_f1);
  plans.exec_r2c(d_s2, d_f2);
  srtb_hip::check_hip(
      srtb_hip::correlate_pointwise(d_f1, d_f2, d_f1, nullptr, nc,
                                    1.0f / (float)count, st),
      "corr");
  plans.exec_c2c_backward(d_f1);
  srtb_hip::check_hip(
      srtb_hip::complex_abs(d_f1, d_mag, nc, st), "abs");
  std::vector<float> h_out(nc);
  srtb_hip::check_hip(hipMemcpyAsync(h_out.data(), d_mag, nc * sizeof(float),
                                     hipMemcpyDeviceToHost, st),
                      "d2h");
  srtb_hip::check_hip(hipStreamSynchronize(st), "sync");

  std::ofstream of(out_path, std::ios::binary);
  of.write(reinterpret_cast<const char*>(h_out.data()),
           (std::streamsize)(nc * sizeof(float)));
  size_t peak = 0;
  for (size_t i = 1; i < nc; ++i)
    if (h_out[i] > h_out[peak]) peak = i;
  std::printf("[srtb-correlator] wrote %s (%zu float32); peak index %zu, "
              "value %.3e\n",
              out_path.c_str(), nc, peak, h_out[peak]);
  return 0;
}
