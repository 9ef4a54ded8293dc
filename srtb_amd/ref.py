"""NumPy reference implementations of every device operation.

These are the *oracles*: each HIP/CDNA4 kernel in ``csrc/kernels`` is tested
against the function of the same name here, and the CPU pipeline path
(:mod:`srtb_amd.pipeline.cpu`) is built from them.  Semantics mirror the
reference implementation (file:line cited per function) so that detection
thresholds and output products transfer; the GPU code is an independent
MI355X-native design that must only agree numerically.

All functions are pure; complex spectra are complex64 unless stated otherwise.
FFT conventions follow cuFFT/hipFFT (forward unscaled, backward unscaled) —
NumPy's ``ifft`` is multiplied back by ``n``.
"""

from __future__ import annotations

import numpy as np

# Dispersion constant, MHz^2 pc^-1 cm^3 s
# (reference: userspace/include/srtb/coherent_dedispersion.hpp:67 — the
# "accurate" value 4.148808e3, not tempo2's 4.149378e3)
D_DISPERSION = 4.148808e3

# ---------------------------------------------------------------------------
# FFT windows (reference: userspace/include/srtb/fft/fft_window.hpp:27-110)
# ---------------------------------------------------------------------------


def window_coefficients(kind: str, n: int, dtype=np.float32) -> np.ndarray:
    """Window coefficient table: coef[i] = w(i / (n-1)), i in [0, n).

    kind: "rectangle" (default in the reference), "hann", "hamming".
    cosine-sum form: w(x) = sum_k (-1)^k a_k cos(2 pi k x).
    """
    x = np.arange(n, dtype=np.float64) / max(n - 1, 1)
    if kind == "rectangle":
        w = np.ones(n, dtype=np.float64)
    elif kind == "hann":
        w = 0.5 - 0.5 * np.cos(2 * np.pi * x)
    elif kind == "hamming":
        w = 25.0 / 46.0 - 21.0 / 46.0 * np.cos(2 * np.pi * x)
    else:
        raise ValueError(f"unknown window {kind!r}")
    return w.astype(dtype)


# ---------------------------------------------------------------------------
# Unpack (reference: userspace/include/srtb/unpack.hpp:43-403)
# ---------------------------------------------------------------------------


def unpack(data: np.ndarray, nbits: int, window: np.ndarray | None = None,
           dtype=np.float32) -> np.ndarray:
    """Unpack packed baseband bytes to float samples.

    ``nbits``: 1, 2, 4 → unsigned sub-byte fields, MSB-first within each byte
    (reference unpack.hpp:43-140); 8 → uint8 cast; -8 → int8 cast; 16/-16,
    32/-32 → u/int casts (unpack.hpp:143-156).  ``window`` (optional) is
    multiplied element-wise (the reference fuses the FFT window here).
    """
    data = np.asarray(data)
    if nbits in (1, 2, 4):
        b = np.frombuffer(data.tobytes(), dtype=np.uint8)
        per = 8 // nbits
        shifts = np.arange(per - 1, -1, -1, dtype=np.uint8) * nbits
        mask = (1 << nbits) - 1
        out = ((b[:, None] >> shifts[None, :]) & mask).reshape(-1).astype(dtype)
    elif abs(nbits) == 8:
        b = np.frombuffer(data.tobytes(), dtype=np.int8 if nbits < 0 else np.uint8)
        out = b.astype(dtype)
    elif abs(nbits) == 16:
        b = np.frombuffer(data.tobytes(), dtype=np.int16 if nbits < 0 else np.uint16)
        out = b.astype(dtype)
    elif abs(nbits) == 32:
        b = np.frombuffer(data.tobytes(), dtype=np.int32 if nbits < 0 else np.uint32)
        out = b.astype(dtype)
    else:
        raise ValueError(f"unsupported nbits {nbits}")
    if window is not None:
        out = (out * window[: out.size]).astype(dtype)
    return out


def unpack_interleaved_2pol(data: np.ndarray, window: np.ndarray | None = None,
                            dtype=np.float32) -> tuple[np.ndarray, np.ndarray]:
    """int8 samples interleaved sample-by-sample: p0 s0, p1 s0, p0 s1, ...

    (reference unpack.hpp:221-244).  Returns (pol0, pol1)."""
    b = np.frombuffer(np.asarray(data).tobytes(), dtype=np.int8).astype(dtype)
    p0, p1 = b[0::2].copy(), b[1::2].copy()
    if window is not None:
        p0 = (p0 * window[: p0.size]).astype(dtype)
        p1 = (p1 * window[: p1.size]).astype(dtype)
    return p0, p1


def unpack_naocpsr_snap1(data: np.ndarray, window: np.ndarray | None = None,
                         dtype=np.float32) -> tuple[np.ndarray, np.ndarray]:
    """SNAP-1 "1 1 2 2" interleave: 2 int8 samples per pol alternating
    (reference unpack.hpp:255-283).  Returns (pol0, pol1)."""
    b = np.frombuffer(np.asarray(data).tobytes(), dtype=np.int8)
    b = b.reshape(-1, 4)  # [s0p0, s1p0, s0p1, s1p1]
    p0 = b[:, 0:2].reshape(-1).astype(dtype)
    p1 = b[:, 2:4].reshape(-1).astype(dtype)
    if window is not None:
        p0 = (p0 * window[: p0.size]).astype(dtype)
        p1 = (p1 * window[: p1.size]).astype(dtype)
    return p0, p1


def unpack_gznupsr_a1(data: np.ndarray, n_streams: int = 2,
                      window: np.ndarray | None = None,
                      dtype=np.float32) -> list[np.ndarray]:
    """GZNU ZCU111 4-byte-word deinterleave (reference unpack.hpp:291-403).

    Packet payload is a sequence of 4-byte words cycling over ``n_streams``
    ADCs; each word holds 4 consecutive samples of one stream.  The 4-stream
    (v1) variant stores offset-binary bytes, fixed by XOR 0x80; the 2-stream
    (v2, current) variant is plain int8.
    """
    b = np.frombuffer(np.asarray(data).tobytes(), dtype=np.uint8)
    words = b.reshape(-1, n_streams, 4)
    outs = []
    for s in range(n_streams):
        v = words[:, s, :].reshape(-1)
        if n_streams == 4:
            v = v ^ np.uint8(0x80)
        v = v.astype(np.int8).astype(dtype)
        if window is not None:
            v = (v * window[: v.size]).astype(dtype)
        outs.append(v)
    return outs


# ---------------------------------------------------------------------------
# FFT stages (library in the GPU path; NumPy here, cuFFT scaling convention)
# ---------------------------------------------------------------------------


def fft_r2c_drop_nyquist(x: np.ndarray) -> np.ndarray:
    """Forward R2C of the full block, dropping the Nyquist bin so the output
    count is exactly N/2 (reference fft_pipe.hpp:77 'drop the highest
    frequency point')."""
    X = np.fft.rfft(np.asarray(x, dtype=np.float64))
    return X[:-1].astype(np.complex64)


def waterfall_ifft(spec: np.ndarray, n_channels: int) -> np.ndarray:
    """Batched backward C2C over contiguous chunks: the Nc-bin dedispersed
    spectrum is viewed as [n_channels][L] (L = Nc / n_channels contiguous fine
    bins per coarse channel) and each row is inverse-FFT'd to L time samples
    (reference watfft_1d_c2c_pipe, fft_pipe.hpp:294-311).  cuFFT backward is
    unscaled, hence the * L."""
    spec = np.asarray(spec)
    nc = spec.size
    L = nc // n_channels
    m = spec.reshape(n_channels, L)
    out = np.fft.ifft(m, axis=1) * L
    return out.astype(np.complex64)


# ---------------------------------------------------------------------------
# RFI mitigation stage 1 (reference: pipeline/rfi_mitigation_pipe.hpp:50-101,
# spectrum/rfi_mitigation.hpp:42-157)
# ---------------------------------------------------------------------------


def rfi_mitigate_s1(spec: np.ndarray, threshold: float,
                    spectrum_channel_count: int) -> np.ndarray:
    """Zap bins with |X|^2 > threshold * mean(|X|^2); scale survivors by
    (Nc^2 / S)^(-1/2) (normalization fused in, rfi_mitigation_pipe.hpp:60-80)."""
    spec = np.asarray(spec)
    n = spec.size
    power = (spec.real.astype(np.float64)) ** 2 + (spec.imag.astype(np.float64)) ** 2
    avg = power.mean()
    coeff = (float(n) * float(n) / float(spectrum_channel_count)) ** -0.5
    out = np.where(power > threshold * avg, 0.0, spec * coeff)
    return out.astype(spec.dtype)


def rfi_ranges_to_bins(freq_low: float, bandwidth: float, n_bins: int,
                       ranges: list[tuple[float, float]]) -> list[tuple[int, int]]:
    """Map RFI frequency ranges (MHz) to inclusive bin index ranges
    (reference spectrum/rfi_mitigation.hpp:97-157; bin i sits at
    freq_low + bandwidth * i / (n_bins - 1); handles negative bandwidth)."""
    out = []
    bw_neg = bandwidth < 0
    for lo, hi in ranges:
        if (hi - lo < 0) != bw_neg:
            lo, hi = hi, lo
        i_lo = int(round((lo - freq_low) / bandwidth * (n_bins - 1)))
        i_hi = int(round((hi - freq_low) / bandwidth * (n_bins - 1)))
        if 0 <= i_lo <= i_hi < n_bins:
            out.append((i_lo, i_hi))
        # else: out of band -> warn and skip (reference logs a warning)
    return out


def parse_rfi_freq_list(text: str) -> list[tuple[float, float]]:
    """Parse "11-12, 15-90" style lists (reference eval_rfi_ranges)."""
    ranges = []
    for part in text.split(","):
        part = part.strip()
        if not part:
            continue
        nums = [p for p in part.split("-") if p != ""]
        if len(nums) != 2:
            continue  # reference logs a warning and skips
        ranges.append((float(nums[0]), float(nums[1])))
    return ranges


def rfi_mitigate_manual(spec: np.ndarray, freq_low: float, bandwidth: float,
                        ranges: list[tuple[float, float]]) -> np.ndarray:
    spec = np.asarray(spec).copy()
    for i_lo, i_hi in rfi_ranges_to_bins(freq_low, bandwidth, spec.size, ranges):
        spec[i_lo : i_hi + 1] = 0
    return spec


# ---------------------------------------------------------------------------
# Coherent dedispersion (reference: coherent_dedispersion.hpp:40-248)
# ---------------------------------------------------------------------------


def dedisp_phase_factors(n: int, f_min: float, f_c: float, df: float,
                         dm: float) -> np.ndarray:
    """Phase factor per frequency bin, computed in float64 like the reference's
    phase_factor_v3: k = D*1e6 * dm / f * ((f-f_c)/f_c)^2 (cycles; may be ~1e9),
    factor = exp(-2*pi*i*frac(k)).  f = f_min + df*i, frequencies in MHz."""
    i = np.arange(n, dtype=np.float64)
    f = f_min + df * i
    delta_f = f - f_c
    k = (D_DISPERSION * 1e6) * dm / f * (delta_f / f_c) ** 2
    k_frac = k - np.trunc(k)  # C modf keeps the sign, like trunc
    delta_phi = -2.0 * np.pi * k_frac
    return (np.cos(delta_phi) + 1j * np.sin(delta_phi)).astype(np.complex64)


def coherent_dedisperse(spec: np.ndarray, f_min: float, f_c: float, df: float,
                        dm: float) -> np.ndarray:
    spec = np.asarray(spec)
    factors = dedisp_phase_factors(spec.size, f_min, f_c, df, dm)
    return (spec * factors).astype(spec.dtype)


def dispersion_delay_time(f: float, f_c: float, dm: float) -> float:
    """Delay (s) of frequency f relative to f_c (MHz), positive when f > f_c
    (reference coherent_dedispersion.hpp:76-79)."""
    return -D_DISPERSION * dm * (1.0 / (f * f) - 1.0 / (f_c * f_c))


def nsamps_reserved(baseband_input_count: int, spectrum_channel_count: int,
                    freq_low: float, bandwidth: float, sample_rate: float,
                    dm: float, reserve: bool = True) -> int:
    """Overlap (in real samples) between adjacent blocks so dedispersion edge
    garbage can be dropped; the valid region is rounded down to a multiple of
    2 * spectrum_channel_count (reference coherent_dedispersion.hpp:87-128)."""
    if not reserve:
        return 0
    max_delay = dispersion_delay_time(freq_low + bandwidth, freq_low, dm)
    minimal = 2 * round(max_delay * sample_rate)
    per_bin = 2 * spectrum_channel_count
    refft_total = (baseband_input_count - minimal) // per_bin * per_bin
    n_may = baseband_input_count - refft_total
    if refft_total > 0:
        return int(n_may)
    return 0  # reference warns and disables overlap


# ---------------------------------------------------------------------------
# Spectral kurtosis RFI (stage 2)
# (reference: spectrum/rfi_mitigation.hpp:183-340)
# ---------------------------------------------------------------------------


def spectral_kurtosis_sk(wf: np.ndarray) -> np.ndarray:
    """SK statistic per frequency row of a [n_channels][M] waterfall:
    SK = M * S4 / S2^2 with S2 = sum |x|^2, S4 = sum |x|^4."""
    wf = np.asarray(wf)
    p = (wf.real.astype(np.float64)) ** 2 + (wf.imag.astype(np.float64)) ** 2
    s2 = p.sum(axis=1)
    s4 = (p * p).sum(axis=1)
    M = wf.shape[1]
    with np.errstate(divide="ignore", invalid="ignore"):
        sk = M * s4 / (s2 * s2)
    return sk


def rfi_mitigate_sk(wf: np.ndarray, sk_threshold: float) -> np.ndarray:
    """Method 2: zero whole frequency rows whose SK is outside the corrected
    [lo', hi'] band; lo = 2 - thr, hi = thr, x' = x*(M-1)/(M+1) + 1
    (reference rfi_mitigation.hpp:292-340)."""
    wf = np.asarray(wf).copy()
    M = wf.shape[1]
    hi = float(sk_threshold)
    lo = 2.0 - hi
    if lo > hi:
        lo, hi = hi, lo
    corr = (M - 1.0) / (M + 1.0)
    lo_, hi_ = lo * corr + 1.0, hi * corr + 1.0
    sk = spectral_kurtosis_sk(wf)
    zap = (sk > hi_) | (sk < lo_)
    wf[zap, :] = 0
    return wf


def rfi_mitigate_sk_v1(wf_tf: np.ndarray, sk_threshold: float,
                       normalize: bool = False) -> np.ndarray:
    """Method 1 (reference rfi_mitigation.hpp:183-274): TIME-MAJOR layout
    [M][fft_bins] (one thread per frequency bin walks the M time samples of
    its column), same corrected SK thresholds as method 2, zeroing whole
    frequency columns; optional per-column normalization by sqrt(mean |x|^2)
    of the surviving data."""
    wf = np.asarray(wf_tf).copy()
    M, bins = wf.shape
    hi = float(sk_threshold)
    lo = 2.0 - hi
    if lo > hi:
        lo, hi = hi, lo
    corr = (M - 1.0) / (M + 1.0)
    lo_, hi_ = lo * corr + 1.0, hi * corr + 1.0
    p = wf.real.astype(np.float64) ** 2 + wf.imag.astype(np.float64) ** 2
    s2 = p.sum(axis=0)
    s4 = (p * p).sum(axis=0)
    with np.errstate(divide="ignore", invalid="ignore"):
        sk = M * s4 / (s2 * s2)
    zap = (sk > hi_) | (sk < lo_) | (s2 == 0)
    wf[:, zap] = 0
    if normalize:
        mp = np.where(zap, 1.0, s2 / M)
        scale = np.where(zap, 0.0, 1.0 / np.sqrt(mp))
        wf = (wf * scale[None, :]).astype(wf.dtype)
    return wf


# ---------------------------------------------------------------------------
# Signal detection (reference: pipeline/signal_detect_pipe.hpp:252-441,
# signal_detect.hpp:25-70)
# ---------------------------------------------------------------------------


def zapped_channel_count(wf: np.ndarray) -> int:
    """Count channels whose FIRST time sample has zero power (zapped rows all
    start with 0; reference signal_detect_pipe.hpp:261-281 samples column 0
    via a stride-permutation iterator)."""
    wf = np.asarray(wf)
    col0 = wf[:, 0]
    p = col0.real.astype(np.float64) ** 2 + col0.imag.astype(np.float64) ** 2
    return int((p == 0).sum())


def time_series_sum(wf: np.ndarray, time_series_count: int,
                    dtype=np.float32) -> np.ndarray:
    """ts[j] = sum over channels of |wf[i][j]|^2, j < time_series_count
    (reference signal_detect_pipe.hpp:305-316).  float32 accumulation to match
    the GPU kernel."""
    wf = np.asarray(wf)
    p = (wf.real.astype(dtype)) ** 2 + (wf.imag.astype(dtype)) ** 2
    return p[:, :time_series_count].sum(axis=0, dtype=dtype)


def count_signal(ts: np.ndarray, snr_threshold: float) -> tuple[int, float]:
    """Threshold = snr * sqrt(mean(ts^2)) (ts assumed zero-mean), return
    (#above, threshold) (reference signal_detect.hpp:33-67)."""
    ts = np.asarray(ts, dtype=np.float64)
    thr = snr_threshold * np.sqrt((ts * ts).mean())
    return int((ts > thr).sum()), float(thr)


def boxcar_series(ts: np.ndarray, boxcar_length: int) -> np.ndarray:
    """box[i] = cumsum[i + L] - cumsum[i], inclusive scan with init 0
    (= sum of ts[i+1 .. i+L]); output length len(ts) - L
    (reference signal_detect_pipe.hpp:374-423)."""
    ts = np.asarray(ts)
    cum = np.cumsum(ts, dtype=np.float64)  # inclusive scan: cum[i] = sum ts[0..i]
    L = boxcar_length
    n = ts.size - L
    return (cum[L : L + n] - cum[0:n]).astype(ts.dtype)


def detect_signals(wf: np.ndarray, nsamps_reserved_real: int,
                   snr_threshold: float, channel_threshold: float,
                   max_boxcar_length: int) -> dict:
    """Full detection stage on a [n_channels][M] waterfall.

    Returns dict with 'time_series' (baseline-subtracted), 'zero_count', and
    'detections': list of (boxcar_length, signal_count, series) — boxcar 1 is
    the raw series.  Mirrors reference signal_detect_pipe_2.
    """
    wf = np.asarray(wf)
    n_channels, m = wf.shape
    zero_count = zapped_channel_count(wf)
    time_reserved = nsamps_reserved_real // n_channels
    ts_count = m if m <= time_reserved else m - time_reserved
    ts = time_series_sum(wf, ts_count)
    ts = (ts - ts.mean(dtype=np.float64)).astype(ts.dtype)

    detections = []
    if zero_count < channel_threshold * n_channels:
        cnt, thr = count_signal(ts, snr_threshold)
        if cnt > 0:
            detections.append((1, cnt, ts.copy()))
        L = 2
        while L <= max_boxcar_length and L < ts.size:
            box = boxcar_series(ts, L)
            cnt, thr = count_signal(box, snr_threshold)
            if cnt > 0:
                detections.append((L, cnt, box))
            L *= 2
    return {"time_series": ts, "zero_count": zero_count,
            "detections": detections}


# ---------------------------------------------------------------------------
# Spectrum simplification for display
# (reference: spectrum/simplify_spectrum.hpp:37-731)
# ---------------------------------------------------------------------------


def resample_power_2d(wf_power: np.ndarray, out_h: int, out_w: int) -> np.ndarray:
    """Area-averaged resample of a [n_channels][M] power waterfall to
    [out_h][out_w] — the capability of resample_spectrum_3 (v2 math: full
    average over the covered source region with fractional edge coverage,
    simplify_spectrum.hpp:276-620)."""
    src = np.asarray(wf_power, dtype=np.float64)
    h, w = src.shape

    def overlap_matrix(n_out, n_in):
        # W[o, i] = length of overlap between out-cell o (width n_in/n_out in
        # source units) and source cell i, normalized so each row sums to 1.
        W = np.zeros((n_out, n_in))
        step = n_in / n_out
        for o in range(n_out):
            a, b = o * step, (o + 1) * step
            i0, i1 = int(np.floor(a)), int(np.ceil(b))
            for i in range(i0, min(i1, n_in)):
                W[o, i] = max(0.0, min(b, i + 1) - max(a, i))
            W[o] /= step
        return W

    return overlap_matrix(out_h, h) @ src @ overlap_matrix(out_w, w).T


def normalize_by_mean(img: np.ndarray) -> np.ndarray:
    """x *= 1 / (2 * mean) (reference simplify_spectrum.hpp:627-644)."""
    img = np.asarray(img, dtype=np.float64)
    m = img.mean()
    if m == 0:
        return img
    return img / (2.0 * m)


# GUI colors (reference config.hpp:60-68)
COLOR_0 = 0xFF1F1E33
COLOR_1 = 0xFF33E1F1
COLOR_OVERFLOW = 0xFFE0E1CC


def generate_pixmap(intensity: np.ndarray) -> np.ndarray:
    """intensity in [0,1] → ARGB32 via per-channel lerp between COLOR_0 and
    COLOR_1; out-of-range → COLOR_OVERFLOW (reference simplify_spectrum.hpp:700-731)."""
    x = np.asarray(intensity, dtype=np.float64)
    out = np.empty(x.shape, dtype=np.uint32)
    ok = (x >= 0) & (x <= 1)

    def chan(c0, c1):
        return np.clip(np.round(c0 + (c1 - c0) * x), 0, 255).astype(np.uint32)

    a = chan((COLOR_0 >> 24) & 0xFF, (COLOR_1 >> 24) & 0xFF)
    r = chan((COLOR_0 >> 16) & 0xFF, (COLOR_1 >> 16) & 0xFF)
    g = chan((COLOR_0 >> 8) & 0xFF, (COLOR_1 >> 8) & 0xFF)
    b = chan(COLOR_0 & 0xFF, COLOR_1 & 0xFF)
    out[:] = (a << 24) | (r << 16) | (g << 8) | b
    out[~ok] = COLOR_OVERFLOW
    return out


# ---------------------------------------------------------------------------
# Misc device algorithms
# ---------------------------------------------------------------------------


def running_mean_init_average(data: np.ndarray, windowsize: int) -> np.ndarray:
    """Per-channel mean of the first ``windowsize`` time samples of a
    time-major [nsamp][nchan] array (reference algorithm/running_mean.hpp:61-77)."""
    x = np.asarray(data, dtype=np.float64)
    return x[:windowsize, :].sum(axis=0) / windowsize


def running_mean(data: np.ndarray, windowsize: int,
                 ave: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
    """1-bit threshold of a time-major [nsamp][nchan] array against a
    sliding-window running mean (reference algorithm/running_mean.hpp:31-59).

    out[t][j] = (data[t][j] > ave_j) where ave_j slides forward by
    (data[t+window][j] - data[t][j]) / window; the last ``window`` outputs
    update the mean from a mirrored tail.  Returns (out, updated_ave).
    """
    x = np.asarray(data, dtype=np.float64)
    nsamp, nchan = x.shape
    out = np.zeros((nsamp, nchan), dtype=np.uint8)
    ave = np.asarray(ave, dtype=np.float64).copy()
    for j in range(nchan):
        a = ave[j]
        for i in range(windowsize, nsamp):
            head = x[i - windowsize, j]
            tail = x[i, j]
            out[i - windowsize, j] = 1 if head > a else 0
            a += (tail - head) / windowsize
        for i in range(windowsize):
            head = x[nsamp + i - windowsize, j]
            tail = x[nsamp - i - 1, j]
            out[i + nsamp - windowsize, j] = 1 if head > a else 0
            a += (tail - head) / windowsize
        ave[j] = a
    return out, ave


def correlate_spectra(f1: np.ndarray, f2: np.ndarray, scale: float) -> np.ndarray:
    """corr[i] = scale * f1[i] * conj(f2[i]) (reference src/correlator.cpp:116-119)."""
    f1 = np.asarray(f1)
    f2 = np.asarray(f2)
    return (scale * f1 * np.conj(f2)).astype(f1.dtype)
