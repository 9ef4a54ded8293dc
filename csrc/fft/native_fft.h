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
#include <cmath>
#include <stdexcept>
#include <string>
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
  // maxcol_log2_ovr / final_log2_ovr (0 = policy default) override the
  // factorization defaults for THIS plan — the engine plans its backward
  // (preop-fused) FFT with 32-max columns + final 256: the fp64
  // dedispersion fused into the first pass wants the 6-wave/SIMD pair32
  // kernel, not the 4-wave pair64 (r02 bench profile: 8.96 vs 4.56
  // ms/block contended).  Explicit SRTB_FFT_* env still wins.
  void plan(size_t len, size_t batch, int sign, hipStream_t stream,
            int maxcol_log2_ovr = 0, int final_log2_ovr = 0) {
    destroy();
    if (!supported(len)) throw std::runtime_error("NativeFft: unsupported len");
    len_ = len;
    batch_ = batch;
    sign_ = sign;

    const size_t L = len;
    // Decomposition (see csrc/kernels/fft.hip):
    //   t = log2(len) <= 12           -> single LDS Stockham pass
    //   t > 12: column factors (register kernel, N in {2..64}) + final 256
    //           (in-place DIF in LDS, digit-reversal folded into the store)
    const int t = ilog2z(len);
    passes_.clear();
    if (t <= 12) {
      // n = 4096 runs as ONE in-place radix-4 DIF pass (n_prefix = 0 →
      // the kernel's digit addressing degenerates to row*n): the ping-pong
      // Stockham at that length needs >160 KB for F=2, so it runs F=1 at
      // 1 WG/CU — the in-place DIF at 65 KB gets 2 WG/CU and measured
      // 1.46x faster (1.07 vs 1.56 ms on 2^27 elements).  At 256/1024 the
      // Stockham kernel measured FASTER than the DIF (0.69/0.77 vs
      // 0.85/0.84) and keeps the job.
      // wave-local kernel for 256/512/1024 (one FFT per wave, no
      // barriers on the data path): measured 1.15-1.37x over the
      // ping-pong Stockham (0.517/0.544/0.673 vs 0.698/0.746/0.771 ms on
      // 2^27 elements, r02 check6).  SRTB_FFT_WAVE=0 reverts.
      const char* we = std::getenv("SRTB_FFT_WAVE");
      // 256-2048: measured 0.515/0.528/0.598/1.291 ms vs Stockham
      // 0.698/0.746/0.771/1.42 (r02 checks 6-8); SRTB_FFT_WAVE=0 reverts
      const int wv = we ? std::atoi(we) : 1;
      if (t >= 8 && t <= 11 && wv != 0) {
        ensure_len_table((uint32_t)len, sign, stream);
        Pass p;
        p.kind = PassKind::kWave;
        p.d.n = (uint32_t)len;
        p.n_ffts = batch;
        p.tw_n = len_table((uint32_t)len);
        passes_.push_back(p);
        return;
      }
      if (t == 12 && !std::getenv("SRTB_FFT_NOSP_DIF")) {
        ensure_len_table((uint32_t)len, sign, stream);
        Pass p;
        p.kind = PassKind::kDif;
        p.d.n = (uint32_t)len;
        p.dif.n = (uint32_t)len;
        p.dif.out_c2 = L;
        p.dif.out_elem_coef = 1;
        p.dif.n_prefix = 0;
        p.n_ffts = batch;
        p.tw_n = len_table((uint32_t)len);
        passes_.push_back(p);
        return;
      }
      ensure_len_table((uint32_t)len, sign, stream);
      Pass p;
      p.kind = PassKind::kStockham;
      p.d.n = (uint32_t)len;
      p.d.d0 = 0;  // sentinel: q0 = id (contiguous rows)
      p.d.in_c0 = L;
      p.d.in_stride = 1;
      p.d.out_c0 = L;
      p.d.out_stride = 1;
      p.n_ffts = batch;
      p.load_ffast = p.store_ffast = false;
      p.tw_n = len_table((uint32_t)len);
      passes_.push_back(p);
      return;
    }
    std::vector<uint32_t> f;
    // SRTB_FFT_FACTORS="64,64,64,8,256": explicit factorization override
    // (all factors incl. the final DIF length; product must equal len) —
    // for plan-shape experiments without planner changes
    if (const char* fe = std::getenv("SRTB_FFT_FACTORS")) {
      size_t prod = 1;
      std::string str(fe);
      for (size_t pos = 0; pos < str.size();) {
        size_t comma = str.find(',', pos);
        if (comma == std::string::npos) comma = str.size();
        const int v = std::atoi(str.substr(pos, comma - pos).c_str());
        if (v > 1) { f.push_back((uint32_t)v); prod *= (size_t)v; }
        pos = comma + 1;
      }
      if (prod != len || f.size() < 2 || f.size() > 5)
        f.clear();  // invalid override for this length: fall through
      else {
        for (size_t i = 0; i + 1 < f.size(); ++i)
          if (f[i] > 64 && f[i] != 512) f.clear();  // 512 = LDS mid pass
      }
    }
    if (!f.empty()) {
      // accepted override
    } else {
      const char* mc = std::getenv("SRTB_FFT_MAXCOL");
      const int maxcol_log2 =
          mc ? ilog2z(std::atoi(mc))
             : (maxcol_log2_ovr ? maxcol_log2_ovr : 6);  // default 64
      const char* fe = std::getenv("SRTB_FFT_FINAL");
      // default final DIF length 64 (3 radix-4 stages, F=32 dif -> 17 KB
      // LDS, 8 WG/CU): fwd 2^29 8.89 -> 8.28 ms, bwd 2^18x2048 5.11 ->
      // 4.89 (r02 dif64 sweep).  t=13 keeps 256 (one fewer pass there).
      int final_log2 =
          fe ? ilog2z(std::atoi(fe))
             : (final_log2_ovr ? final_log2_ovr : (t >= 14 ? 6 : 8));
      if (final_log2 & 1) ++final_log2;                 // pure 4^t only
      if (final_log2 < 6) final_log2 = 6;
      if (final_log2 > 12) final_log2 = 12;
      if (final_log2 >= t) final_log2 = (t % 2) ? t - 1 : t - 2;
      const int rest = t - final_log2;
      if (rest >= 18 && rest <= 24) {
        // large residual: greedy 64s + one remainder factor.  The N=64
        // lane-pair kernel is the fastest pass per 8 GB sweep (4 waves/SIMD,
        // measured 2.2 vs 2.75 ms under contention), and 18..20 bits save a
        // whole pass vs balanced 32s: 2^29 fwd [64,64,64,8] measured 9.41 ms
        // vs [64,32,32,32] 9.88 (fft_factor_sweep.py).
        int left = rest;
        while (left >= 6 && (int)f.size() < 4) {
          f.push_back(64);
          left -= 6;
        }
        if (left > 0) f.push_back(1u << left);
      } else {
        // small residual: distribute evenly over 32-max columns (measured
        // best at the 2^18 waterfall: [32,32] beats [64,16])
        int ncols = (rest + maxcol_log2 - 1) / maxcol_log2;
        if (ncols > 4) ncols = 4;  // widen factors instead (up to 64 each)
        const int base_b = rest / ncols, extra = rest % ncols;
        for (int i = 0; i < ncols; ++i) {
          const int b = base_b + (i < extra ? 1 : 0);
          if (b > 6) throw std::runtime_error("NativeFft: factor too large");
          f.push_back(1u << b);
        }
      }
      f.push_back(1u << final_log2);
    }
    const int m = (int)f.size();
    if (m - 1 > 4) throw std::runtime_error("NativeFft: too many factors");

    // suffix products M[j] = prod(f[j..m-1]); M[m] = 1
    std::vector<size_t> M(m + 1, 1);
    for (int j = m - 1; j >= 0; --j) M[j] = M[j + 1] * f[j];
    // prefix products C[i] = prod(f[0..i-1])
    std::vector<size_t> Cp(m + 1, 1);
    for (int i = 1; i <= m; ++i) Cp[i] = Cp[i - 1] * f[i - 1];

    for (int j = 0; j + 1 < m; ++j) {  // column passes
      ensure_len_table(f[j], sign, stream);
      ensure_mod_table(M[j], sign, stream);
      Pass p;
      p.kind = (f[j] == 512) ? PassKind::kMid : PassKind::kCol;
      p.d.n = f[j];
      p.d.d0 = (uint32_t)M[j + 1];
      p.d.d1 = (uint32_t)Cp[j];
      p.d.in_c0 = 1;
      p.d.in_c1 = M[j];
      p.d.in_c2 = L;
      p.d.in_stride = M[j + 1];
      p.d.tw_f0 = 1;
      p.d.tw_mod = M[j];
      p.d.tw_lo_bits = mod_lo_bits(M[j]);
      p.d.tw_angle = sign * 2.0 * M_PI / (double)M[j];
      p.n_ffts = batch * (L / f[j]);
      p.tw_n = len_table(f[j]);
      p.tw_hi = mod_hi(M[j]);
      p.tw_lo = mod_lo(M[j]);
      passes_.push_back(p);
    }
    {  // final DIF pass
      const uint32_t fn = f[m - 1];
      ensure_len_table(fn, sign, stream);
      Pass p;
      p.kind = PassKind::kDif;
      p.d.n = fn;
      p.dif.n = fn;
      p.dif.out_c2 = L;
      p.dif.out_elem_coef = Cp[m - 1];  // prod of all prefix factors
      p.dif.n_prefix = m - 1;
      for (int w = 0; w < m - 1; ++w) {
        // id low bits = k_0 (output-order digits); stored-prefix coefficient
        // S_w = prod(f_{w+1} .. f_{m-2}) = M[w+1] / fn
        p.dif.pf_bits[w] = ilog2z(f[w]);
        p.dif.pf_coef[w] = M[w + 1] / fn;
      }
      p.n_ffts = batch * (L / fn);
      p.tw_n = len_table(fn);
      passes_.push_back(p);
    }
  }

  int n_passes() const { return (int)passes_.size(); }

  // The engine can fuse the RFI+dedispersion elementwise stage into the
  // first pass when that pass is a register column pass.
  bool first_pass_fusable() const {
    return !passes_.empty() && passes_[0].kind == PassKind::kCol;
  }

  // SK-stat accumulation can fuse into the final DIF pass when each
  // workgroup stays within one batch row; returns workgroups per row
  // (partials buffer = batch * wgs_per_row float2), or 0 if unsupported.
  int dif_sk_wgs_per_row() const {
    if (passes_.empty() || passes_.back().kind != PassKind::kDif) return 0;
    const Pass& p = passes_.back();
    const size_t per_row = (len_ / p.dif.n);  // instances per batch row
    const int F = dif_f(p);
    if (per_row % F != 0) return 0;
    return (int)(per_row / F);
  }

  // Execute the planned transform.  out may equal in only for 1-pass plans.
  // preop (optional) is applied to the FIRST pass's loads (requires
  // first_pass_fusable()).
  // first_pass_out (optional): write the FIRST column pass out-of-place
  // into this buffer (later passes continue in place there).  REQUIRED
  // when preop->r2c_m is set: the pair-combine reads element m-k of the
  // input, which an in-place pass would be overwriting concurrently.
  void exec(const float2* in, float2* out, hipStream_t stream,
            const FftPreop* preop = nullptr,
            float2* dif_sk_partials = nullptr,
            const uint8_t* decode2_raw = nullptr, int decode_bits = 2,
            float2* first_pass_out = nullptr) {
    if (passes_.empty()) throw std::runtime_error("NativeFft: not planned");
    if (passes_.size() > 1 && in == out && !first_pass_out)
      throw std::runtime_error("NativeFft: multi-pass needs out != in");
    if (preop && preop->r2c_m && !first_pass_out)
      throw std::runtime_error("NativeFft: r2c fusion needs first_pass_out");
    if (preop && !first_pass_fusable())
      throw std::runtime_error("NativeFft: preop needs a column first pass");
    if (decode2_raw && (preop || !first_pass_fusable()))
      throw std::runtime_error("NativeFft: decode2 needs a column first pass");
    float2* cur = const_cast<float2*>(in);
    for (size_t i = 0; i < passes_.size(); ++i) {
      Pass& p = passes_[i];
      const bool last = (i + 1 == passes_.size());
      float2* dst = last ? out : cur;
      const FftPreop* pre = (i == 0) ? preop : nullptr;
      switch (p.kind) {
        case PassKind::kStockham: {
          const int F = pick_f(p.d.n, p, p.n_ffts);
          check_hip(fft_stockham_pass(cur, dst, p.d, p.n_ffts, F,
                                      p.load_ffast, p.store_ffast, sign_,
                                      p.tw_n, p.tw_hi, p.tw_lo, stream),
                    "fft_stockham_pass");
          break;
        }
        case PassKind::kCol: {
          float2* o = (i == 0 && first_pass_out) ? first_pass_out : cur;
          check_hip(fft_col_pass(cur, o, p.d, p.n_ffts, sign_, p.tw_n,
                                 p.tw_hi, p.tw_lo, stream, pre,
                                 i == 0 ? decode2_raw : nullptr, decode_bits),
                    "fft_col_pass");
          dst = o;  // in place except an out-of-place first pass
          break;
        }
        case PassKind::kMid:
          check_hip(fft_mid512_pass(cur, cur, p.d, p.n_ffts, sign_, p.tw_n,
                                    stream),
                    "fft_mid512_pass");
          dst = cur;  // in place
          break;
        case PassKind::kWave:
          check_hip(fft_wave_pass(cur, dst, p.d.n, p.n_ffts, sign_, p.tw_n,
                                  stream),
                    "fft_wave_pass");
          break;
        case PassKind::kDif: {
          const int F = dif_f(p);
          check_hip(fft_dif_final(cur, dst, p.dif, p.n_ffts, F, sign_,
                                  p.tw_n, dif_sk_partials, stream),
                    "fft_dif_final");
          break;
        }
      }
      cur = dst;
    }
  }

  void destroy() {
    for (auto& t : tables_) (void)hipFree(t.ptr);
    tables_.clear();
    passes_.clear();
  }

 private:
  enum class PassKind { kStockham, kCol, kMid, kDif, kWave };

  struct Pass {
    PassKind kind = PassKind::kStockham;
    FftPassDesc d{};
    DifFinalDesc dif{};
    size_t n_ffts = 0;
    bool load_ffast = false, store_ffast = false;
    const float2* tw_n = nullptr;
    const float2* tw_hi = nullptr;
    const float2* tw_lo = nullptr;
  };

  struct Table {
    int kind;      // 0 = per-length butterfly, 1 = mod_hi, 2 = mod_lo
    size_t key;    // length or modulus
    float2* ptr;
  };

  static int ilog2z(size_t v) {
    int t = 0;
    while ((1ull << t) < v) ++t;
    return t;
  }

  int dif_f(const Pass& p) const {
    // target <= 80 KiB LDS so 2 workgroups fit per CU (the mid512 lesson:
    // occupancy dominates run length on these latency-bound LDS kernels).
    // SRTB_FFT_DIF_F overrides (r02 PMC: the DIF pass runs 2.7-3.1 TB/s
    // vs 5.2-5.6 for the column passes, 38% wave-park — occupancy sweep).
    // F=16 measured fastest (r02 sweep: fwd 2^29 8.89/9.25/9.38/12.96 ms
    // and bwd 2^18x2048 5.11/5.40/6.01/7.77 at F=16/8/32/4): 35 KB LDS ->
    // 4 workgroups/CU overlap the per-stage barriers that parked 38% of
    // wave cycles at F=32 (see profiles/r02_pmc_summary.md).
    int F = (p.dif.n <= 64) ? 32 : 16;
    if (const char* e = std::getenv("SRTB_FFT_DIF_F")) {
      const int v = std::atoi(e);
      if (v >= 1 && v <= 64 && (v & (v - 1)) == 0) F = v;
    }
    while (F > 1 &&
           ((size_t)p.dif.n + (size_t)F * (p.dif.n + 2)) * sizeof(float2) >
               80 * 1024)
      F >>= 1;
    while (F > 1 && p.n_ffts % F != 0) F >>= 1;
    return F;
  }

  int pick_f(uint32_t n, const Pass& p, size_t n_ffts) const {
    size_t f = (size_t)elems_per_wg() / n;
    if (f < 1) f = 1;
    // keep LDS under 160 KiB: (n + 2*F*(n+2) + 16) * 8 (tw tables + ping-pong)
    while (f > 1 &&
           ((size_t)n + 2ull * f * (n + 2) + 16) * sizeof(float2) >
               160 * 1024)
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
