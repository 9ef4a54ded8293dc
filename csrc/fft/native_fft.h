// Hand-written FFT planning for the srtb_amd engine (gfx950).
//
// Plans pow2 C2C transforms as 1-3 Stockham passes (each pass length <=
// kMaxPassLen) with fused inter-pass twiddles and transpose-free output
// scatter; real-input forward transforms use the packed-complex trick +
// r2c_post_process.  Index math: srtb_amd/fftref.py (fft_deep).
//
// Buffer discipline: every pass except the last is in-place safe (a pass
// writes exactly the locations it read, workgroup-locally staged through
// LDS); only the LAST pass scatters, so exec(in, out) runs passes 0..k-2
// in place on `in` and the final pass in -> out (out must differ from in
// unless k == 1).

#pragma once

#include <algorithm>
#include <cstdint>
#include <cstdlib>
#include <stdexcept>
#include <vector>

#include "../include/srtb_kernels.h"
#include "fft_plans.h"  // check_hip

namespace srtb_hip {

class NativeFft {
 public:
  static constexpr uint32_t kMaxPassLen = 4096;
  // elements per workgroup: smaller -> more blocks/CU (latency hiding),
  // larger -> longer coalesced runs on strided passes.  Sweepable via env.
  static int elems_per_wg() {
    const char* e = std::getenv("SRTB_FFT_ELEMS");
    int v = e ? std::atoi(e) : 4096;
    if (v < 512) v = 512;
    if (v > 8192) v = 8192;
    return v;
  }

  static bool is_pow2(size_t v) { return v && (v & (v - 1)) == 0; }

  // supported: pow2 length factorable into <= 3 passes
  static bool supported(size_t len) {
    return is_pow2(len) && len >= 2 &&
           len <= (size_t)kMaxPassLen * kMaxPassLen * kMaxPassLen;
  }

  NativeFft() = default;
  NativeFft(const NativeFft&) = delete;
  NativeFft& operator=(const NativeFft&) = delete;
  ~NativeFft() { destroy(); }

  // Plan a batched C2C: `batch` rows of length `len`, contiguous.
  void plan(size_t len, size_t batch, int sign, hipStream_t stream) {
    destroy();
    if (!supported(len)) throw std::runtime_error("NativeFft: unsupported len");
    len_ = len;
    batch_ = batch;
    sign_ = sign;

    std::vector<uint32_t> f = factorize(len);
    const size_t L = len;

    // twiddle tables per distinct pass length
    for (uint32_t fl : f) ensure_len_table(fl, sign, stream);

    passes_.clear();
    if (f.size() == 1) {
      Pass p;
      p.d.n = f[0];
      p.d.d0 = 0;  // sentinel: q0 = id (contiguous rows; batch not pow2-safe)
      p.d.in_c0 = L;  // row stride per instance (q0 = global row)
      p.d.in_stride = 1;
      p.d.out_c0 = L;
      p.d.out_stride = 1;
      p.n_ffts = batch;
      p.load_ffast = false;
      p.store_ffast = false;
      p.contig_rows = true;
      p.tw_n = len_table(f[0]);
      passes_.push_back(p);
    } else if (f.size() == 2) {
      const uint32_t f0 = f[0], f1 = f[1];
      ensure_mod_table((size_t)f0 * f1, sign, stream);
      {  // pass 1: columns (len f0, stride f1), twiddle mod L
        Pass p;
        p.d.n = f0;
        p.d.d0 = f1; p.d.d1 = 1;
        p.d.in_c0 = 1; p.d.in_c2 = L; p.d.in_stride = f1;
        p.d.out_c0 = 1; p.d.out_c2 = L; p.d.out_stride = f1;
        p.d.tw_f0 = 1; p.d.tw_mod = L; p.d.tw_lo_bits = mod_lo_bits(L);
        p.n_ffts = batch * f1;
        p.load_ffast = p.store_ffast = true;
        p.tw_n = len_table(f0);
        p.tw_hi = mod_hi(L); p.tw_lo = mod_lo(L);
        passes_.push_back(p);
      }
      {  // pass 2: rows (len f1, contiguous), scatter out[k1 + f0*k2]
        Pass p;
        p.d.n = f1;
        p.d.d0 = f0; p.d.d1 = 1;
        p.d.in_c0 = f1; p.d.in_c2 = L; p.d.in_stride = 1;
        p.d.out_c0 = 1; p.d.out_c2 = L; p.d.out_stride = f0;
        p.n_ffts = batch * f0;
        p.load_ffast = false;
        p.store_ffast = true;
        p.tw_n = len_table(f1);
        passes_.push_back(p);
      }
    } else {  // 3 factors
      const uint32_t f0 = f[0], f1 = f[1], f2 = f[2];
      const size_t f12 = (size_t)f1 * f2;
      ensure_mod_table(L, sign, stream);
      ensure_mod_table(f12, sign, stream);
      {  // pass A: len f0, stride f1*f2, twiddle (q0=n_hat)*k0 mod L
        Pass p;
        p.d.n = f0;
        p.d.d0 = (uint32_t)f12; p.d.d1 = 1;
        p.d.in_c0 = 1; p.d.in_c2 = L; p.d.in_stride = f12;
        p.d.out_c0 = 1; p.d.out_c2 = L; p.d.out_stride = f12;
        p.d.tw_f0 = 1; p.d.tw_mod = L; p.d.tw_lo_bits = mod_lo_bits(L);
        p.n_ffts = batch * f12;
        p.load_ffast = p.store_ffast = true;
        p.tw_n = len_table(f0);
        p.tw_hi = mod_hi(L); p.tw_lo = mod_lo(L);
        passes_.push_back(p);
      }
      {  // pass B: len f1, stride f2, within each [k0] chunk; tw mod f1*f2
        Pass p;
        p.d.n = f1;
        p.d.d0 = f2; p.d.d1 = f0;
        p.d.in_c0 = 1; p.d.in_c1 = f12; p.d.in_c2 = L; p.d.in_stride = f2;
        p.d.out_c0 = 1; p.d.out_c1 = f12; p.d.out_c2 = L; p.d.out_stride = f2;
        p.d.tw_f0 = 1; p.d.tw_mod = f12; p.d.tw_lo_bits = mod_lo_bits(f12);
        p.n_ffts = batch * f0 * f2;
        p.load_ffast = p.store_ffast = true;
        p.tw_n = len_table(f1);
        p.tw_hi = mod_hi(f12); p.tw_lo = mod_lo(f12);
        passes_.push_back(p);
      }
      {  // pass C: len f2, contiguous rows (k0,k1); scatter k0 + f0*k1 + f0f1*k2
        Pass p;
        p.d.n = f2;
        p.d.d0 = f0; p.d.d1 = f1;
        p.d.in_c0 = f12; p.d.in_c1 = f2; p.d.in_c2 = L; p.d.in_stride = 1;
        p.d.out_c0 = 1; p.d.out_c1 = f0; p.d.out_c2 = L;
        p.d.out_stride = (size_t)f0 * f1;
        p.n_ffts = batch * f0 * f1;
        p.load_ffast = false;
        p.store_ffast = true;
        p.tw_n = len_table(f2);
        passes_.push_back(p);
      }
    }
  }

  int n_passes() const { return (int)passes_.size(); }

  // Execute the planned transform.  out may equal in only for 1-pass plans.
  void exec(const float2* in, float2* out, hipStream_t stream) {
    if (passes_.empty()) throw std::runtime_error("NativeFft: not planned");
    if (passes_.size() > 1 && in == out)
      throw std::runtime_error("NativeFft: multi-pass needs out != in");
    float2* cur = const_cast<float2*>(in);
    for (size_t i = 0; i < passes_.size(); ++i) {
      Pass& p = passes_[i];
      const bool last = (i + 1 == passes_.size());
      float2* dst = last ? out : cur;
      FftPassDesc d = p.d;
      size_t n_ffts = p.n_ffts;
      const int F = pick_f(d.n, p, n_ffts);
      check_hip(fft_stockham_pass(cur, dst, d, n_ffts, F, p.load_ffast,
                                  p.store_ffast, sign_, p.tw_n, p.tw_hi,
                                  p.tw_lo, stream),
                "fft_stockham_pass");
      cur = dst;
    }
  }

  void destroy() {
    for (auto& t : tables_) hipFree(t.ptr);
    tables_.clear();
    passes_.clear();
  }

 private:
  struct Pass {
    FftPassDesc d{};
    size_t n_ffts = 0;
    bool load_ffast = false, store_ffast = false;
    bool contig_rows = false;
    const float2* tw_n = nullptr;
    const float2* tw_hi = nullptr;
    const float2* tw_lo = nullptr;
  };

  struct Table {
    int kind;      // 0 = per-length butterfly, 1 = mod_hi, 2 = mod_lo
    size_t key;    // length or modulus
    float2* ptr;
  };

  static std::vector<uint32_t> factorize(size_t len) {
    // split into <= 3 balanced pow2 factors, each <= kMaxPassLen;
    // the LAST factor is the largest (it gets the contiguous-load pass).
    if (len <= kMaxPassLen) return {(uint32_t)len};
    int t = 0;
    while ((1ull << t) < len) ++t;
    const int tm = 12;  // log2(kMaxPassLen)
    if (len <= (size_t)kMaxPassLen * kMaxPassLen) {
      int t2 = std::min(tm, (t + 1) / 2);
      int t1 = t - t2;
      if (t1 > tm) { t1 = tm; t2 = t - tm; }
      return {(uint32_t)(1u << std::min(t1, t2)),
              (uint32_t)(1u << std::max(t1, t2))};
    }
    int t3 = std::min(tm, (t + 2) / 3);
    int rem = t - t3;
    int t2 = std::min(tm, (rem + 1) / 2);
    int t1 = rem - t2;
    if (t1 > tm) throw std::runtime_error("NativeFft: length too large");
    std::vector<int> v{t1, t2, t3};
    std::sort(v.begin(), v.end());
    return {(uint32_t)(1u << v[0]), (uint32_t)(1u << v[1]),
            (uint32_t)(1u << v[2])};
  }

  int pick_f(uint32_t n, const Pass& p, size_t n_ffts) const {
    size_t f = (size_t)elems_per_wg() / n;
    if (f < 1) f = 1;
    // keep LDS under 160 KiB: (n + 2*F*(n+2)) * 8  (tw table + ping-pong)
    while (f > 1 &&
           ((size_t)n + 2ull * f * (n + 2)) * sizeof(float2) > 160 * 1024)
      f >>= 1;
    while (f > 1 && n_ffts % f != 0) f >>= 1;
    // instances in a workgroup must share q1/q2 digits only if... they need
    // not; addressing is exact per instance.  But f-fast coalescing wants
    // q0-runs: cap f at d0 when the pass is strided.
    if (p.d.d0 > 1 && f > p.d.d0) f = p.d.d0;
    while (f > 1 && n_ffts % f != 0) f >>= 1;
    return (int)f;
  }

  void ensure_len_table(uint32_t n, int sign, hipStream_t stream) {
    if (len_table(n)) return;
    float2* t = nullptr;
    // FULL circle: radix-4 stages index up to 3n/4
    check_hip(hipMalloc(&t, std::max<size_t>(n, 1) * sizeof(float2)),
              "tw_n alloc");
    check_hip(fft_build_twiddle(t, std::max<size_t>(n, 1), (double)n, sign,
                                stream),
              "tw_n build");
    tables_.push_back({0, n, t});
  }

  static int mod_lo_bits_for(size_t mod) {
    int t = 0;
    while ((1ull << t) < mod) ++t;
    return t / 2;
  }

  void ensure_mod_table(size_t mod, int sign, hipStream_t stream) {
    if (mod_hi(mod)) return;
    const int lo_bits = mod_lo_bits_for(mod);
    const size_t lo_n = 1ull << lo_bits;
    const size_t hi_n = mod >> lo_bits;
    float2 *thi = nullptr, *tlo = nullptr;
    check_hip(hipMalloc(&thi, hi_n * sizeof(float2)), "tw_hi alloc");
    check_hip(hipMalloc(&tlo, lo_n * sizeof(float2)), "tw_lo alloc");
    // hi[j] = exp(s*2πi*j*2^lo/mod) = exp(s*2πi*j/hi_n)
    check_hip(fft_build_twiddle(thi, hi_n, (double)hi_n, sign, stream),
              "tw_hi build");
    check_hip(fft_build_twiddle(tlo, lo_n, (double)mod, sign, stream),
              "tw_lo build");
    tables_.push_back({1, mod, thi});
    tables_.push_back({2, mod, tlo});
  }

  float2* len_table(uint32_t n) const {
    for (auto& t : tables_)
      if (t.kind == 0 && t.key == n) return t.ptr;
    return nullptr;
  }
  float2* mod_hi(size_t mod) const {
    for (auto& t : tables_)
      if (t.kind == 1 && t.key == mod) return t.ptr;
    return nullptr;
  }
  float2* mod_lo(size_t mod) const {
    for (auto& t : tables_)
      if (t.kind == 2 && t.key == mod) return t.ptr;
    return nullptr;
  }
  int mod_lo_bits(size_t mod) const { return mod_lo_bits_for(mod); }

  size_t len_ = 0, batch_ = 0;
  int sign_ = -1;
  std::vector<Pass> passes_;
  std::vector<Table> tables_;
};

}  // namespace srtb_hip
