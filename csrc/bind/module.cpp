// Torch-extension bindings for srtb_amd: granular kernel ops (for numerics
// tests and composition) + the native PipelineEngine.
//
// The kernels themselves live in csrc/kernels/*.hip (pure HIP, no torch);
// this TU only adapts torch::Tensor ↔ raw pointers and streams.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>

#include <functional>
#include <vector>

#include "../engine/engine.h"
#include "../fft/native_fft.h"
#include "../include/srtb_kernels.h"

namespace {

using namespace srtb_hip;

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

void check(hipError_t e, const char* what) { check_hip(e, what); }

#define CHECK_CUDA(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHECK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

float2* cptr(torch::Tensor& t) {
  return reinterpret_cast<float2*>(t.data_ptr());
}
const float2* cptr(const torch::Tensor& t) {
  return reinterpret_cast<const float2*>(t.data_ptr());
}

torch::Tensor t_unpack(torch::Tensor raw, int64_t nbits, int64_t out_count,
                       c10::optional<torch::Tensor> window) {
  CHECK_CUDA(raw);
  CHECK_CONTIG(raw);
  auto out = torch::empty({out_count},
                          raw.options().dtype(torch::kFloat32));
  const float* w = nullptr;
  if (window.has_value()) {
    CHECK_CUDA(*window);
    w = window->data_ptr<float>();
  }
  check(unpack(raw.data_ptr<uint8_t>(), out.data_ptr<float>(), out_count,
               (int)nbits, w, cur_stream()),
        "unpack");
  return out;
}

std::vector<torch::Tensor> t_unpack_2pol(torch::Tensor raw, std::string kind) {
  CHECK_CUDA(raw);
  CHECK_CONTIG(raw);
  const size_t per_pol = raw.numel() / 2;
  auto o0 = torch::empty({(int64_t)per_pol}, raw.options().dtype(torch::kFloat32));
  auto o1 = torch::empty({(int64_t)per_pol}, raw.options().dtype(torch::kFloat32));
  auto* p = reinterpret_cast<const int8_t*>(raw.data_ptr());
  if (kind == "interleave")
    check(unpack_interleaved_2pol(p, o0.data_ptr<float>(), o1.data_ptr<float>(),
                                  per_pol, nullptr, cur_stream()),
          "unpack_2pol");
  else if (kind == "naocpsr_snap1")
    check(unpack_naocpsr_snap1(p, o0.data_ptr<float>(), o1.data_ptr<float>(),
                               per_pol, nullptr, cur_stream()),
          "unpack_snap1");
  else
    TORCH_CHECK(false, "unknown 2-pol kind ", kind);
  return {o0, o1};
}

std::vector<torch::Tensor> t_unpack_gznupsr(torch::Tensor raw,
                                            int64_t n_streams) {
  CHECK_CUDA(raw);
  CHECK_CONTIG(raw);
  const size_t per = raw.numel() / n_streams;
  std::vector<torch::Tensor> outs;
  float* ptrs[4] = {nullptr, nullptr, nullptr, nullptr};
  for (int i = 0; i < n_streams; ++i) {
    outs.push_back(torch::empty({(int64_t)per},
                                raw.options().dtype(torch::kFloat32)));
    ptrs[i] = outs.back().data_ptr<float>();
  }
  check(unpack_gznupsr_a1(raw.data_ptr<uint8_t>(), ptrs[0], ptrs[1], ptrs[2],
                          ptrs[3], (int)n_streams, per, nullptr, cur_stream()),
        "unpack_gznupsr");
  return outs;
}

torch::Tensor t_mean_power(torch::Tensor spec) {
  CHECK_CUDA(spec);
  CHECK_CONTIG(spec);
  TORCH_CHECK(spec.scalar_type() == torch::kComplexFloat);
  auto partials = torch::empty({reduce_partials()},
                               spec.options().dtype(torch::kFloat64));
  auto out = torch::empty({1}, spec.options().dtype(torch::kFloat64));
  check(mean_power(cptr(spec), spec.numel(),
                   partials.data_ptr<double>(), out.data_ptr<double>(),
                   cur_stream()),
        "mean_power");
  return out;
}

torch::Tensor t_sum_sumsq(torch::Tensor x) {
  CHECK_CUDA(x);
  CHECK_CONTIG(x);
  auto partials = torch::empty({reduce_partials()},
                               x.options().dtype(torch::kFloat64));
  auto out = torch::empty({2}, x.options().dtype(torch::kFloat64));
  check(sum_sumsq(x.data_ptr<float>(), x.numel(), partials.data_ptr<double>(),
                  out.data_ptr<double>(), cur_stream()),
        "sum_sumsq");
  return out;
}

void t_rfi_s1(torch::Tensor spec, double threshold, int64_t channel_count) {
  CHECK_CUDA(spec);
  CHECK_CONTIG(spec);
  auto mean = t_mean_power(spec);
  const size_t n = spec.numel();
  const float coeff =
      (float)std::pow((double)n * (double)n / (double)channel_count, -0.5);
  check(rfi_s1(cptr(spec), n, mean.data_ptr<double>(), (float)threshold,
               coeff, cur_stream()),
        "rfi_s1");
}

void t_zap_bins(torch::Tensor spec, int64_t lo, int64_t hi) {
  CHECK_CUDA(spec);
  check(zap_bins(cptr(spec), lo, hi, cur_stream()), "zap_bins");
}

void t_dedisperse(torch::Tensor spec, double f_min, double f_c, double df,
                  double dm) {
  CHECK_CUDA(spec);
  CHECK_CONTIG(spec);
  check(dedisperse(cptr(spec), spec.numel(), f_min, f_c, df, dm, cur_stream()),
        "dedisperse");
}

torch::Tensor t_phase_table(int64_t n, double f_min, double f_c, double df,
                            double dm, torch::Device dev) {
  auto out = torch::empty({n}, torch::TensorOptions()
                                   .dtype(torch::kComplexFloat)
                                   .device(dev));
  check(dedisp_phase_table(cptr(out), n, f_min, f_c, df, dm, cur_stream()),
        "phase_table");
  return out;
}

void t_rfi_dedisperse_fused(torch::Tensor spec, bool enable_rfi,
                            double threshold, int64_t channel_count,
                            std::vector<std::vector<int64_t>> ranges,
                            double f_min, double f_c, double df, double dm,
                            c10::optional<torch::Tensor> table) {
  CHECK_CUDA(spec);
  CHECK_CONTIG(spec);
  const size_t n = spec.numel();
  torch::Tensor mean;
  const double* mp = nullptr;
  if (enable_rfi) {
    mean = t_mean_power(spec);
    mp = mean.data_ptr<double>();
  }
  const float coeff =
      (float)std::pow((double)n * (double)n / (double)channel_count, -0.5);
  ZapRange zr[16];
  TORCH_CHECK(ranges.size() <= 16, "at most 16 zap ranges");
  for (size_t i = 0; i < ranges.size(); ++i) {
    zr[i].lo = (unsigned long long)ranges[i][0];
    zr[i].hi = (unsigned long long)ranges[i][1];
  }
  const float2* tb = nullptr;
  if (table.has_value()) tb = cptr(*table);
  check(rfi_dedisperse_fused(cptr(spec), n, mp, (float)threshold, coeff, zr,
                             (int)ranges.size(), f_min, f_c, df, dm, tb,
                             cur_stream()),
        "rfi_dedisperse_fused");
}

torch::Tensor t_sk_row_stats(torch::Tensor wf) {
  CHECK_CUDA(wf);
  CHECK_CONTIG(wf);
  TORCH_CHECK(wf.dim() == 2);
  const size_t rows = wf.size(0), len = wf.size(1);
  auto out = torch::empty({(int64_t)rows, 2},
                          wf.options().dtype(torch::kFloat32));
  check(sk_row_stats(cptr(wf), rows, len,
                     reinterpret_cast<float2*>(out.data_ptr<float>()),
                     cur_stream()),
        "sk_row_stats");
  return out;
}

std::vector<torch::Tensor> t_sk_mitigate(torch::Tensor wf, double sk_threshold) {
  CHECK_CUDA(wf);
  CHECK_CONTIG(wf);
  TORCH_CHECK(wf.dim() == 2);
  const size_t rows = wf.size(0), len = wf.size(1);
  auto s2s4 = t_sk_row_stats(wf);
  double hi = sk_threshold, lo = 2.0 - hi;
  if (lo > hi) std::swap(lo, hi);
  const double corr = ((double)len - 1.0) / ((double)len + 1.0);
  auto flags = torch::empty({(int64_t)rows}, wf.options().dtype(torch::kUInt8));
  auto zero_count = torch::zeros({1}, wf.options().dtype(torch::kInt32));
  check(sk_flags(cptr(wf), reinterpret_cast<float2*>(s2s4.data_ptr<float>()),
                 rows, len, (float)(lo * corr + 1.0), (float)(hi * corr + 1.0),
                 flags.data_ptr<uint8_t>(),
                 reinterpret_cast<unsigned*>(zero_count.data_ptr<int32_t>()),
                 cur_stream()),
        "sk_flags");
  check(sk_zap_rows(cptr(wf), flags.data_ptr<uint8_t>(), rows, len,
                    cur_stream()),
        "sk_zap");
  return {flags, zero_count};
}

torch::Tensor t_sk_v1_mitigate(torch::Tensor wf, double sk_threshold,
                               bool normalize) {
  CHECK_CUDA(wf);
  CHECK_CONTIG(wf);
  TORCH_CHECK(wf.dim() == 2);  // [M][bins], time-major
  const size_t M = wf.size(0), bins = wf.size(1);
  auto s2s4 = torch::empty({(int64_t)bins, 2},
                           wf.options().dtype(torch::kFloat32));
  check(sk_v1_stats(cptr(wf), M, bins,
                    reinterpret_cast<float2*>(s2s4.data_ptr<float>()),
                    cur_stream()),
        "sk_v1_stats");
  double hi = sk_threshold, lo = 2.0 - hi;
  if (lo > hi) std::swap(lo, hi);
  const double corr = ((double)M - 1.0) / ((double)M + 1.0);
  check(sk_v1_zap(cptr(wf), M, bins,
                  reinterpret_cast<const float2*>(s2s4.data_ptr<float>()),
                  (float)(lo * corr + 1.0), (float)(hi * corr + 1.0),
                  normalize, cur_stream()),
        "sk_v1_zap");
  return s2s4;
}

torch::Tensor t_time_series(torch::Tensor wf,
                            c10::optional<torch::Tensor> flags,
                            int64_t ts_count) {
  CHECK_CUDA(wf);
  CHECK_CONTIG(wf);
  TORCH_CHECK(wf.dim() == 2);
  const size_t rows = wf.size(0), len = wf.size(1);
  auto out = torch::empty({ts_count}, wf.options().dtype(torch::kFloat32));
  const uint8_t* f = flags.has_value() ? flags->data_ptr<uint8_t>() : nullptr;
  auto scratch = torch::empty(
      {(int64_t)time_series_chunks(ts_count) * ts_count},
      wf.options().dtype(torch::kFloat32));
  check(time_series_2stage(cptr(wf), f, rows, len, ts_count,
                           out.data_ptr<float>(), scratch.data_ptr<float>(),
                           cur_stream()),
        "time_series");
  return out;
}

void t_subtract_mean(torch::Tensor ts) {
  CHECK_CUDA(ts);
  auto sums = t_sum_sumsq(ts);
  check(subtract_mean(ts.data_ptr<float>(), ts.numel(),
                      sums.data_ptr<double>(), cur_stream()),
        "subtract_mean");
}

std::vector<torch::Tensor> t_count_signal(torch::Tensor ts, double snr) {
  CHECK_CUDA(ts);
  auto sums = t_sum_sumsq(ts);
  auto count = torch::zeros({1}, ts.options().dtype(torch::kInt32));
  auto thr = torch::empty({1}, ts.options().dtype(torch::kFloat32));
  check(count_above(ts.data_ptr<float>(), ts.numel(),
                    sums.data_ptr<double>() + 1, (float)snr,
                    reinterpret_cast<unsigned*>(count.data_ptr<int32_t>()),
                    thr.data_ptr<float>(), cur_stream()),
        "count_above");
  return {count, thr};
}

torch::Tensor t_inclusive_scan(torch::Tensor ts) {
  CHECK_CUDA(ts);
  auto out = torch::empty_like(ts);
  auto scratch = torch::empty({std::max<int64_t>(4096,
                                  scan_scratch_size(ts.numel()))},
                              ts.options());
  check(inclusive_scan(ts.data_ptr<float>(), out.data_ptr<float>(),
                       ts.numel(), scratch.data_ptr<float>(), cur_stream()),
        "inclusive_scan");
  return out;
}

torch::Tensor t_boxcar(torch::Tensor cumsum, int64_t L) {
  CHECK_CUDA(cumsum);
  const int64_t n_out = cumsum.numel() - L;
  auto out = torch::empty({n_out}, cumsum.options());
  check(boxcar(cumsum.data_ptr<float>(), out.data_ptr<float>(), n_out, L,
               cur_stream()),
        "boxcar");
  return out;
}

torch::Tensor t_resample_power(torch::Tensor wf, int64_t H, int64_t W) {
  CHECK_CUDA(wf);
  CHECK_CONTIG(wf);
  TORCH_CHECK(wf.dim() == 2);
  auto out = torch::empty({H, W}, wf.options().dtype(torch::kFloat32));
  check(resample_power_2d(cptr(wf), wf.size(0), wf.size(1),
                          out.data_ptr<float>(), (int)H, (int)W, cur_stream()),
        "resample");
  return out;
}

void t_normalize_by_mean(torch::Tensor img) {
  CHECK_CUDA(img);
  auto sums = t_sum_sumsq(img);
  check(normalize_by_mean(img.data_ptr<float>(), img.numel(),
                          sums.data_ptr<double>(), cur_stream()),
        "normalize");
}

torch::Tensor t_generate_pixmap(torch::Tensor img, int64_t c0, int64_t c1,
                                int64_t cover) {
  CHECK_CUDA(img);
  auto out = torch::empty_like(img, img.options().dtype(torch::kInt32));
  check(generate_pixmap(img.data_ptr<float>(),
                        reinterpret_cast<uint32_t*>(out.data_ptr<int32_t>()),
                        img.numel(), (uint32_t)c0, (uint32_t)c1,
                        (uint32_t)cover, cur_stream()),
        "pixmap");
  return out;
}

std::vector<torch::Tensor> t_running_mean(torch::Tensor data,
                                          int64_t windowsize) {
  CHECK_CUDA(data);
  CHECK_CONTIG(data);
  TORCH_CHECK(data.dim() == 2);
  const size_t nsamp = data.size(0), nchan = data.size(1);
  auto ave = torch::empty({(int64_t)nchan}, data.options());
  auto out = torch::empty({(int64_t)nsamp, (int64_t)nchan},
                          data.options().dtype(torch::kUInt8));
  check(running_mean_init(data.data_ptr<float>(), nsamp, nchan, windowsize,
                          ave.data_ptr<float>(), cur_stream()),
        "rm init");
  check(running_mean(data.data_ptr<float>(), nsamp, nchan,
                     out.data_ptr<uint8_t>(), windowsize,
                     ave.data_ptr<float>(), cur_stream()),
        "rm");
  return {out, ave};
}

std::vector<torch::Tensor> t_correlate(torch::Tensor f1, torch::Tensor f2,
                                       double scale) {
  CHECK_CUDA(f1);
  CHECK_CUDA(f2);
  auto corr = torch::empty_like(f1);
  auto mag = torch::empty({f1.numel()}, f1.options().dtype(torch::kFloat32));
  check(correlate_pointwise(cptr(f1), cptr(f2), cptr(corr),
                            mag.data_ptr<float>(), f1.numel(), (float)scale,
                            cur_stream()),
        "correlate");
  return {corr, mag};
}

// ---------------- hand-written FFT (test/bench surface) ----------------

torch::Tensor t_native_fft(torch::Tensor x, int64_t sign) {
  CHECK_CUDA(x);
  CHECK_CONTIG(x);
  TORCH_CHECK(x.scalar_type() == torch::kComplexFloat);
  const size_t len = x.size(-1);
  const size_t batch = x.numel() / len;
  auto stream = cur_stream();
  NativeFft plan;
  plan.plan(len, batch, (int)sign, stream);
  auto out = torch::empty_like(x);
  if (plan.n_passes() == 1) {
    plan.exec(cptr(x), cptr(out), stream);
  } else {
    auto work = x.clone();  // passes 0..k-2 run in place on the input copy
    plan.exec(cptr(work), cptr(out), stream);
  }
  check(hipStreamSynchronize(stream), "native_fft sync");  // plan is local
  return out;
}

torch::Tensor t_native_rfft(torch::Tensor x) {
  // real forward via packed-complex trick + r2c post; returns n/2 bins
  CHECK_CUDA(x);
  CHECK_CONTIG(x);
  TORCH_CHECK(x.scalar_type() == torch::kFloat32);
  const size_t n = x.numel();
  const size_t m = n / 2;
  auto stream = cur_stream();
  NativeFft plan;
  plan.plan(m, 1, -1, stream);
  auto z = torch::empty({(int64_t)m},
                        x.options().dtype(torch::kComplexFloat));
  auto packed = x.view({(int64_t)m, 2});  // reinterpret as complex pairs
  auto work = torch::empty_like(z);
  work.copy_(torch::view_as_complex(packed));
  plan.exec(cptr(work), cptr(z), stream);
  check(r2c_post_process(cptr(z), cptr(z), m, nullptr, nullptr, stream),
        "r2c post");
  check(hipStreamSynchronize(stream), "native_rfft sync");
  return z;
}

// ---------------- FFT microbenchmarks (plan once, event-timed) ----------------

double time_iters(hipStream_t stream, int iters, const std::function<void()>& f) {
  hipEvent_t e0, e1;
  check(hipEventCreate(&e0), "ev");
  check(hipEventCreate(&e1), "ev");
  f();  // warmup
  check(hipStreamSynchronize(stream), "warm sync");
  check(hipEventRecord(e0, stream), "rec");
  for (int i = 0; i < iters; ++i) f();
  check(hipEventRecord(e1, stream), "rec");
  check(hipEventSynchronize(e1), "sync");
  float ms = 0;
  check(hipEventElapsedTime(&ms, e0, e1), "elapsed");
  hipEventDestroy(e0);
  hipEventDestroy(e1);
  return ms / iters;
}

double t_bench_fft(int64_t len, int64_t batch, int64_t sign, int64_t iters,
                   std::string backend) {
  auto stream = cur_stream();
  auto x = torch::randn({batch, len, 2},
                        torch::TensorOptions().dtype(torch::kFloat32)
                            .device(torch::kCUDA));
  auto y = torch::empty_like(x);
  float2* xp = reinterpret_cast<float2*>(x.data_ptr());
  float2* yp = reinterpret_cast<float2*>(y.data_ptr());
  double ms = 0;
  if (backend == "native") {
    NativeFft plan;
    plan.plan(len, batch, (int)sign, stream);
    ms = time_iters(stream, (int)iters, [&] { plan.exec(xp, yp, stream); });
  } else {
    hipfftHandle h;
    check_fft(hipfftCreate(&h), "create");
    long long n1[1] = {(long long)len};
    size_t ws = 0;
    check_fft(hipfftMakePlanMany64(h, 1, n1, nullptr, 1, len, nullptr, 1, len,
                                   HIPFFT_C2C, batch, &ws),
              "plan");
    check_fft(hipfftSetStream(h, stream), "stream");
    ms = time_iters(stream, (int)iters, [&] {
      hipfftExecC2C(h, reinterpret_cast<hipfftComplex*>(xp),
                    reinterpret_cast<hipfftComplex*>(yp),
                    sign < 0 ? HIPFFT_FORWARD : HIPFFT_BACKWARD);
    });
    hipfftDestroy(h);
  }
  return ms;
}

double t_bench_rfft(int64_t n, int64_t iters, std::string backend) {
  auto stream = cur_stream();
  auto x = torch::randn({n}, torch::TensorOptions()
                                 .dtype(torch::kFloat32)
                                 .device(torch::kCUDA));
  auto y = torch::empty({n / 2 + 1, 2}, x.options());
  float* xp = x.data_ptr<float>();
  float2* yp = reinterpret_cast<float2*>(y.data_ptr());
  double ms = 0;
  if (backend == "native") {
    const size_t m = n / 2;
    NativeFft plan;
    plan.plan(m, 1, -1, stream);
    ms = time_iters(stream, (int)iters, [&] {
      plan.exec(reinterpret_cast<float2*>(xp), yp, stream);
      check(r2c_post_process(yp, yp, m, nullptr, nullptr, stream), "post");
    });
  } else {
    hipfftHandle h;
    check_fft(hipfftCreate(&h), "create");
    long long n1[1] = {(long long)n};
    size_t ws = 0;
    check_fft(hipfftMakePlanMany64(h, 1, n1, nullptr, 1, 0, nullptr, 1, 0,
                                   HIPFFT_R2C, 1, &ws),
              "plan");
    check_fft(hipfftSetStream(h, stream), "stream");
    ms = time_iters(stream, (int)iters, [&] {
      hipfftExecR2C(h, xp, reinterpret_cast<hipfftComplex*>(yp));
    });
    hipfftDestroy(h);
  }
  return ms;
}

// ---------------- engine binding ----------------

class PyEngine {
 public:
  PyEngine(int64_t n, int64_t nbits, int64_t channels, double freq_low,
           double bandwidth, double sample_rate, double dm,
           double rfi_threshold, double sk_threshold, double snr_threshold,
           int64_t max_boxcar, int64_t nsamps_reserved,
           std::vector<std::vector<int64_t>> zap_ranges, bool use_phase_table,
           bool enable_rfi_s1, bool enable_sk, int64_t n_slots,
           int64_t fft_backend, int64_t window_kind, bool use_hip_graph) {
    EngineConfig c;
    c.baseband_input_count = n;
    c.baseband_input_bits = (int)nbits;
    c.spectrum_channel_count = channels;
    c.freq_low = freq_low;
    c.bandwidth = bandwidth;
    c.sample_rate = sample_rate;
    c.dm = dm;
    c.rfi_threshold = (float)rfi_threshold;
    c.sk_threshold = (float)sk_threshold;
    c.snr_threshold = (float)snr_threshold;
    c.max_boxcar_length = max_boxcar;
    c.nsamps_reserved = nsamps_reserved;
    TORCH_CHECK(zap_ranges.size() <= 16);
    c.n_zap_ranges = (int)zap_ranges.size();
    for (size_t i = 0; i < zap_ranges.size(); ++i) {
      c.zap_ranges[i].lo = (unsigned long long)zap_ranges[i][0];
      c.zap_ranges[i].hi = (unsigned long long)zap_ranges[i][1];
    }
    c.use_phase_table = use_phase_table;
    c.enable_rfi_s1 = enable_rfi_s1;
    c.enable_sk = enable_sk;
    c.fft_backend = (int)fft_backend;
    c.window_kind = (int)window_kind;
    c.use_hip_graph = use_hip_graph;
    eng_ = std::make_unique<PipelineEngine>(c, (int)n_slots);
  }

  // device-tensor submissions: the tensor was produced on torch's current
  // stream while the engine enqueues on its own slot stream — record a
  // producer event and have the slot stream wait on it (no host sync)
  hipEvent_t producer_event() {
    if (!prod_ev_)
      check(hipEventCreateWithFlags(&prod_ev_, hipEventDisableTiming),
            "producer event create");
    check(hipEventRecord(prod_ev_, cur_stream()), "producer event record");
    return prod_ev_;
  }

  int64_t submit(torch::Tensor raw, double dm_override) {
    TORCH_CHECK(raw.is_contiguous());
    TORCH_CHECK((size_t)raw.numel() * raw.element_size() == eng_->raw_bytes(),
                "raw block has wrong byte size");
    if (raw.is_cuda())
      return eng_->submit_device(raw.data_ptr(), eng_->raw_bytes(),
                                 dm_override, producer_event());
    return eng_->submit(raw.data_ptr(), eng_->raw_bytes(), dm_override);
  }

  int64_t submit_samples(torch::Tensor samples, double dm_override) {
    TORCH_CHECK(samples.is_cuda() && samples.is_contiguous());
    TORCH_CHECK(samples.scalar_type() == torch::kFloat32);
    return eng_->submit_samples_device(samples.data_ptr<float>(),
                                       samples.numel(), dm_override,
                                       producer_event());
  }

  py::dict wait(int64_t slot) {
    auto r = eng_->wait((int)slot);
    py::dict d;
    d["zero_count"] = r.zero_count;
    py::list counts;
    for (auto& c : r.counts) counts.append(py::make_tuple(c.first, c.second));
    d["counts"] = counts;
    py::list thr;
    for (float t : r.thresholds) thr.append(t);
    d["thresholds"] = thr;
    return d;
  }

  void synchronize() { eng_->synchronize(); }

  torch::Tensor waterfall(int64_t slot) {
    const auto S = (int64_t)eng_->n_channels(), L = (int64_t)eng_->waterfall_len();
    return torch::from_blob(eng_->waterfall_ptr((int)slot), {S, L},
                            torch::TensorOptions()
                                .dtype(torch::kComplexFloat)
                                .device(torch::kCUDA));
  }

  torch::Tensor time_series(int64_t slot) {
    return torch::from_blob(eng_->time_series_ptr((int)slot),
                            {(int64_t)eng_->ts_count()},
                            torch::TensorOptions()
                                .dtype(torch::kFloat32)
                                .device(torch::kCUDA));
  }

  torch::Tensor boxcar_series(int64_t slot, int64_t L) {
    float* p = eng_->compute_boxcar((int)slot, L);
    auto t = torch::from_blob(p, {(int64_t)(eng_->ts_count() - L)},
                              torch::TensorOptions()
                                  .dtype(torch::kFloat32)
                                  .device(torch::kCUDA));
    return t.clone();
  }

  int64_t n() const { return eng_->n(); }
  int64_t nc() const { return eng_->nc(); }
  int64_t n_channels() const { return eng_->n_channels(); }
  int64_t waterfall_len() const { return eng_->waterfall_len(); }
  int64_t ts_count() const { return eng_->ts_count(); }
  int64_t raw_bytes() const { return eng_->raw_bytes(); }
  int64_t n_slots() const { return eng_->n_slots(); }

 private:
  std::unique_ptr<PipelineEngine> eng_;
  hipEvent_t prod_ev_ = nullptr;

 public:
  ~PyEngine() {
    if (prod_ev_) (void)hipEventDestroy(prod_ev_);
  }
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "srtb_amd MI355X-native kernels + pipeline engine";
  m.def("unpack", &t_unpack, py::arg("raw"), py::arg("nbits"),
        py::arg("out_count"), py::arg("window") = c10::nullopt);
  m.def("unpack_2pol", &t_unpack_2pol);
  m.def("unpack_gznupsr_a1", &t_unpack_gznupsr);
  m.def("mean_power", &t_mean_power);
  m.def("sum_sumsq", &t_sum_sumsq);
  m.def("rfi_s1", &t_rfi_s1);
  m.def("zap_bins", &t_zap_bins);
  m.def("dedisperse", &t_dedisperse);
  m.def("dedisp_phase_table", &t_phase_table);
  m.def("rfi_dedisperse_fused", &t_rfi_dedisperse_fused, py::arg("spec"),
        py::arg("enable_rfi"), py::arg("threshold"), py::arg("channel_count"),
        py::arg("ranges"), py::arg("f_min"), py::arg("f_c"), py::arg("df"),
        py::arg("dm"), py::arg("table") = c10::nullopt);
  m.def("sk_row_stats", &t_sk_row_stats);
  m.def("sk_mitigate", &t_sk_mitigate);
  m.def("sk_v1_mitigate", &t_sk_v1_mitigate, py::arg("wf"),
        py::arg("sk_threshold"), py::arg("normalize") = false);
  m.def("time_series", &t_time_series, py::arg("wf"), py::arg("flags"),
        py::arg("ts_count"));
  m.def("subtract_mean", &t_subtract_mean);
  m.def("count_signal", &t_count_signal);
  m.def("inclusive_scan", &t_inclusive_scan);
  m.def("boxcar", &t_boxcar);
  m.def("resample_power", &t_resample_power);
  m.def("normalize_by_mean", &t_normalize_by_mean);
  m.def("generate_pixmap", &t_generate_pixmap);
  m.def("running_mean", &t_running_mean);
  m.def("correlate", &t_correlate);
  m.def("native_fft", &t_native_fft, py::arg("x"), py::arg("sign"));
  m.def("native_rfft", &t_native_rfft, py::arg("x"));
  m.def("bench_fft", &t_bench_fft, py::arg("len"), py::arg("batch"),
        py::arg("sign"), py::arg("iters") = 20, py::arg("backend") = "native");
  m.def("bench_rfft", &t_bench_rfft, py::arg("n"), py::arg("iters") = 20,
        py::arg("backend") = "native");

  py::class_<PyEngine>(m, "PipelineEngine")
      .def(py::init<int64_t, int64_t, int64_t, double, double, double, double,
                    double, double, double, int64_t, int64_t,
                    std::vector<std::vector<int64_t>>, bool, bool, bool,
                    int64_t, int64_t, int64_t, bool>(),
           py::arg("n"), py::arg("nbits"), py::arg("channels"),
           py::arg("freq_low"), py::arg("bandwidth"), py::arg("sample_rate"),
           py::arg("dm"), py::arg("rfi_threshold") = 10.0,
           py::arg("sk_threshold") = 1.1, py::arg("snr_threshold") = 6.0,
           py::arg("max_boxcar") = 1024, py::arg("nsamps_reserved") = 0,
           py::arg("zap_ranges") = std::vector<std::vector<int64_t>>{},
           py::arg("use_phase_table") = false,
           py::arg("enable_rfi_s1") = true, py::arg("enable_sk") = true,
           py::arg("n_slots") = 2, py::arg("fft_backend") = 2,
           py::arg("window_kind") = 0, py::arg("use_hip_graph") = false)
      .def("submit", &PyEngine::submit, py::arg("raw"),
           py::arg("dm_override") = std::nan(""))
      .def("submit_samples", &PyEngine::submit_samples, py::arg("samples"),
           py::arg("dm_override") = std::nan(""))
      .def("wait", &PyEngine::wait)
      .def("synchronize", &PyEngine::synchronize)
      .def("waterfall", &PyEngine::waterfall)
      .def("time_series", &PyEngine::time_series)
      .def("boxcar_series", &PyEngine::boxcar_series)
      .def_property_readonly("n", &PyEngine::n)
      .def_property_readonly("nc", &PyEngine::nc)
      .def_property_readonly("n_channels", &PyEngine::n_channels)
      .def_property_readonly("waterfall_len", &PyEngine::waterfall_len)
      .def_property_readonly("ts_count", &PyEngine::ts_count)
      .def_property_readonly("raw_bytes", &PyEngine::raw_bytes)
      .def_property_readonly("n_slots", &PyEngine::n_slots);
}
