"""GPU numerics tests: every HIP kernel vs the NumPy/torch fp32 oracle."""

import numpy as np
import pytest
import torch

from srtb_amd import ref

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def C():
    from srtb_amd.ops import native
    torch.cuda.set_device(0)
    return native()


def to_gpu(a):
    return torch.from_numpy(np.ascontiguousarray(a)).cuda()


# ---------------- unpack ----------------

@pytest.mark.parametrize("nbits", [1, 2, 4, 8, -8])
def test_unpack_matches_oracle(C, nbits):
    rng = np.random.default_rng(1)
    raw = rng.integers(0, 256, 1 << 12, dtype=np.uint8)
    expect = ref.unpack(raw, nbits)
    out = C.unpack(to_gpu(raw), nbits, expect.size).cpu().numpy()
    np.testing.assert_array_equal(out, expect)


@pytest.mark.parametrize("nbits", [1, 2, 4])
@pytest.mark.parametrize("n_bytes", [300, 255, 256 + 17])
def test_unpack_subbyte_tail(C, nbits, n_bytes):
    """Non-multiple-of-256-byte inputs take the wave-block main kernel PLUS
    the byte-per-lane tail kernel; pow2-sized tests never exercise the
    seam."""
    rng = np.random.default_rng(10 * nbits + n_bytes)
    raw = rng.integers(0, 256, n_bytes, dtype=np.uint8)
    expect = ref.unpack(raw, nbits)
    out = C.unpack(to_gpu(raw), nbits, expect.size).cpu().numpy()
    np.testing.assert_array_equal(out, expect)
    # and with a window, which the tail kernel must index absolutely
    w = ref.window_coefficients("hann", expect.size)
    expect_w = ref.unpack(raw, nbits, window=w)
    out_w = C.unpack(to_gpu(raw), nbits, expect.size, to_gpu(w)).cpu().numpy()
    np.testing.assert_allclose(out_w, expect_w, rtol=1e-6)


def test_unpack_16bit(C):
    rng = np.random.default_rng(2)
    raw = rng.integers(0, 256, 1 << 12, dtype=np.uint8)
    for b in (16, -16):
        expect = ref.unpack(raw, b)
        out = C.unpack(to_gpu(raw), b, expect.size).cpu().numpy()
        np.testing.assert_array_equal(out, expect)


@pytest.mark.parametrize("nbits", [2, 8, -16, 32])
def test_unpack_with_window(C, nbits):
    """Window fusion must apply to EVERY sample format (the 16/32-bit cast
    cases once hardcoded it off)."""
    rng = np.random.default_rng(3)
    raw = rng.integers(0, 256, 4096, dtype=np.uint8)
    expect_plain = ref.unpack(raw, nbits)
    w = ref.window_coefficients("hamming", expect_plain.size)
    expect = ref.unpack(raw, nbits, window=w)
    out = C.unpack(to_gpu(raw), nbits, expect.size, to_gpu(w)).cpu().numpy()
    np.testing.assert_allclose(out, expect, rtol=1e-5, atol=1e-3)


def test_unpack_2pol_kinds(C):
    rng = np.random.default_rng(4)
    raw = rng.integers(0, 256, 1 << 12, dtype=np.uint8)
    p0, p1 = ref.unpack_interleaved_2pol(raw)
    g0, g1 = C.unpack_2pol(to_gpu(raw), "interleave")
    np.testing.assert_array_equal(g0.cpu().numpy(), p0)
    np.testing.assert_array_equal(g1.cpu().numpy(), p1)
    p0, p1 = ref.unpack_naocpsr_snap1(raw)
    g0, g1 = C.unpack_2pol(to_gpu(raw), "naocpsr_snap1")
    np.testing.assert_array_equal(g0.cpu().numpy(), p0)
    np.testing.assert_array_equal(g1.cpu().numpy(), p1)


@pytest.mark.parametrize("ns", [2, 4])
def test_unpack_gznupsr(C, ns):
    rng = np.random.default_rng(5)
    raw = rng.integers(0, 256, 1 << 12, dtype=np.uint8)
    expect = ref.unpack_gznupsr_a1(raw, n_streams=ns)
    outs = C.unpack_gznupsr_a1(to_gpu(raw), ns)
    for e, o in zip(expect, outs):
        np.testing.assert_array_equal(o.cpu().numpy(), e)


# ---------------- reductions ----------------

def test_mean_power(C):
    rng = np.random.default_rng(6)
    x = (rng.normal(size=1 << 16) + 1j * rng.normal(size=1 << 16)
         ).astype(np.complex64)
    m = C.mean_power(to_gpu(x)).cpu().item()
    expect = np.mean(np.abs(x.astype(np.complex128)) ** 2)
    assert abs(m - expect) < 1e-6 * expect


def test_sum_sumsq(C):
    rng = np.random.default_rng(7)
    x = rng.normal(size=12345).astype(np.float32)
    s = C.sum_sumsq(to_gpu(x)).cpu().numpy()
    np.testing.assert_allclose(s[0], x.astype(np.float64).sum(), rtol=1e-10)
    np.testing.assert_allclose(s[1], (x.astype(np.float64) ** 2).sum(),
                               rtol=1e-10)


# ---------------- RFI s1 / dedispersion ----------------

def test_rfi_s1(C):
    rng = np.random.default_rng(8)
    n, s = 1 << 14, 1 << 8
    spec = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    spec[1000] = 500.0
    expect = ref.rfi_mitigate_s1(spec, 10.0, s)
    g = to_gpu(spec)
    C.rfi_s1(g, 10.0, s)
    np.testing.assert_allclose(g.cpu().numpy(), expect, rtol=2e-5, atol=1e-6)


def test_zap_bins(C):
    spec = np.ones(256, dtype=np.complex64)
    g = to_gpu(spec)
    C.zap_bins(g, 10, 20)
    out = g.cpu().numpy()
    assert (out[10:21] == 0).all()
    assert out[9] == 1 and out[21] == 1


def test_dedisperse_vs_oracle(C):
    rng = np.random.default_rng(9)
    n = 1 << 14
    spec = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    f_min, bw, dm = 1437.0, -64.0, -478.8
    f_c, df = f_min + bw, bw / n
    expect = ref.coherent_dedisperse(spec, f_min, f_c, df, dm)
    g = to_gpu(spec)
    C.dedisperse(g, f_min, f_c, df, dm)
    np.testing.assert_allclose(g.cpu().numpy(), expect, rtol=1e-4, atol=1e-4)


def test_phase_table(C):
    n = 1 << 12
    f_min, bw, dm = 1000.0, 500.0, 478.8
    f_c, df = f_min + bw, bw / n
    expect = ref.dedisp_phase_factors(n, f_min, f_c, df, dm)
    t = C.dedisp_phase_table(n, f_min, f_c, df, dm,
                             torch.device("cuda")).cpu().numpy()
    np.testing.assert_allclose(t, expect, atol=2e-6)


def test_fused_equals_sequential(C):
    rng = np.random.default_rng(10)
    n, s = 1 << 14, 1 << 8
    spec = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    spec[77] = 300.0
    f_min, bw, dm = 1437.0, -64.0, -478.8
    f_c, df = f_min + bw, bw / n
    # sequential oracle
    e = ref.rfi_mitigate_s1(spec, 1.5, s)
    e[100:121] = 0
    e = ref.coherent_dedisperse(e, f_min, f_c, df, dm)
    g = to_gpu(spec)
    C.rfi_dedisperse_fused(g, True, 1.5, s, [[100, 120]], f_min, f_c, df, dm)
    np.testing.assert_allclose(g.cpu().numpy(), e, rtol=1e-4, atol=1e-5)


def test_fused_with_table_matches_fly(C):
    rng = np.random.default_rng(11)
    n, s = 1 << 13, 1 << 7
    spec = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    f_min, bw, dm = 1400.0, 64.0, 100.0
    f_c, df = f_min + bw, bw / n
    g1 = to_gpu(spec)
    g2 = to_gpu(spec)
    tab = C.dedisp_phase_table(n, f_min, f_c, df, dm, torch.device("cuda"))
    C.rfi_dedisperse_fused(g1, True, 5.0, s, [], f_min, f_c, df, dm)
    C.rfi_dedisperse_fused(g2, True, 5.0, s, [], f_min, f_c, df, dm, tab)
    np.testing.assert_allclose(g1.cpu().numpy(), g2.cpu().numpy(), atol=1e-6)


# ---------------- SK ----------------

def test_sk_row_stats(C):
    rng = np.random.default_rng(12)
    wf = (rng.normal(size=(32, 512)) + 1j * rng.normal(size=(32, 512))
          ).astype(np.complex64)
    out = C.sk_row_stats(to_gpu(wf)).cpu().numpy()
    p = np.abs(wf.astype(np.complex128)) ** 2
    np.testing.assert_allclose(out[:, 0], p.sum(axis=1), rtol=1e-4)
    np.testing.assert_allclose(out[:, 1], (p * p).sum(axis=1), rtol=1e-4)


def test_sk_mitigate(C):
    rng = np.random.default_rng(13)
    wf = (rng.normal(size=(32, 2048)) + 1j * rng.normal(size=(32, 2048))
          ).astype(np.complex64)
    wf[7, :] = 0
    wf[7, ::100] = 50.0
    expect = ref.rfi_mitigate_sk(wf, 1.05)
    g = to_gpu(wf)
    flags, zero_count = C.sk_mitigate(g, 1.05)
    out = g.cpu().numpy()
    fl = flags.cpu().numpy()
    assert fl[7] == 1
    assert (out[7] == 0).all()
    ez = (np.abs(expect).sum(axis=1) == 0)
    np.testing.assert_array_equal(fl.astype(bool), ez)
    np.testing.assert_allclose(out, expect, rtol=1e-5)
    assert zero_count.cpu().item() == int(ez.sum())


# ---------------- detection ----------------

def test_time_series(C):
    rng = np.random.default_rng(14)
    wf = (rng.normal(size=(16, 256)) + 1j * rng.normal(size=(16, 256))
          ).astype(np.complex64)
    expect = ref.time_series_sum(wf, 200)
    out = C.time_series(to_gpu(wf), None, 200).cpu().numpy()
    np.testing.assert_allclose(out, expect, rtol=1e-4)


def test_subtract_mean_and_count(C):
    rng = np.random.default_rng(15)
    ts = rng.normal(size=1 << 14).astype(np.float32)
    ts[100] = 50.0
    g = to_gpu(ts)
    C.subtract_mean(g)
    e = ts - ts.mean(dtype=np.float64)
    np.testing.assert_allclose(g.cpu().numpy(), e, atol=1e-4)
    cnt, thr = C.count_signal(g, 6.0)
    ecnt, ethr = ref.count_signal(e.astype(np.float32), 6.0)
    assert cnt.cpu().item() == ecnt
    np.testing.assert_allclose(thr.cpu().item(), ethr, rtol=1e-5)


def test_inclusive_scan(C):
    rng = np.random.default_rng(16)
    for n in (1, 63, 2048, 2049, 1 << 16, (1 << 18) - 7):
        x = rng.normal(size=n).astype(np.float32)
        out = C.inclusive_scan(to_gpu(x)).cpu().numpy()
        expect = np.cumsum(x, dtype=np.float64)
        np.testing.assert_allclose(out, expect, rtol=1e-3, atol=2e-2)


def test_boxcar(C):
    rng = np.random.default_rng(17)
    ts = rng.normal(size=4096).astype(np.float32)
    cum = C.inclusive_scan(to_gpu(ts))
    box = C.boxcar(cum, 8).cpu().numpy()
    expect = ref.boxcar_series(ts, 8)
    np.testing.assert_allclose(box, expect, rtol=1e-3, atol=2e-2)


# ---------------- display ----------------

def test_resample_power(C):
    rng = np.random.default_rng(18)
    wf = (rng.normal(size=(64, 256)) + 1j * rng.normal(size=(64, 256))
          ).astype(np.complex64)
    out = C.resample_power(to_gpu(wf), 16, 32).cpu().numpy()
    expect = ref.resample_power_2d(np.abs(wf.astype(np.complex128)) ** 2, 16, 32)
    np.testing.assert_allclose(out, expect, rtol=1e-4)


def test_normalize_and_pixmap(C):
    rng = np.random.default_rng(19)
    img = rng.random(1000).astype(np.float32)
    g = to_gpu(img)
    C.normalize_by_mean(g)
    expect = ref.normalize_by_mean(img)
    np.testing.assert_allclose(g.cpu().numpy(), expect, rtol=1e-5)
    pix = C.generate_pixmap(g, ref.COLOR_0, ref.COLOR_1,
                            ref.COLOR_OVERFLOW).cpu().numpy().view(np.uint32)
    epix = ref.generate_pixmap(g.cpu().numpy())
    np.testing.assert_array_equal(pix, epix)


def test_running_mean(C):
    rng = np.random.default_rng(20)
    data = rng.random((64, 8)).astype(np.float32)
    out, ave = C.running_mean(to_gpu(data), 16)
    a0 = ref.running_mean_init_average(data, 16)
    eout, eave = ref.running_mean(data, 16, a0)
    np.testing.assert_array_equal(out.cpu().numpy(), eout)
    np.testing.assert_allclose(ave.cpu().numpy(), eave, rtol=1e-4)


def test_correlate(C):
    rng = np.random.default_rng(21)
    f1 = (rng.normal(size=512) + 1j * rng.normal(size=512)).astype(np.complex64)
    f2 = (rng.normal(size=512) + 1j * rng.normal(size=512)).astype(np.complex64)
    corr, mag = C.correlate(to_gpu(f1), to_gpu(f2), 0.25)
    expect = ref.correlate_spectra(f1, f2, 0.25)
    np.testing.assert_allclose(corr.cpu().numpy(), expect, rtol=1e-4, atol=1e-5)
    np.testing.assert_allclose(mag.cpu().numpy(), np.abs(expect), rtol=1e-4,
                               atol=1e-5)


@pytest.mark.parametrize("normalize", [False, True])
def test_sk_v1_mitigate(C, normalize):
    rng = np.random.default_rng(30)
    M, bins = 256, 96  # bins deliberately non-pow2
    wf = (rng.normal(size=(M, bins)) + 1j * rng.normal(size=(M, bins))
          ).astype(np.complex64)
    wf[:, 5] = 0
    wf[::40, 5] = 25.0
    expect = ref.rfi_mitigate_sk_v1(wf, 1.05, normalize=normalize)
    g = to_gpu(wf)
    C.sk_v1_mitigate(g, 1.05, normalize)
    np.testing.assert_allclose(g.cpu().numpy(), expect, rtol=1e-4, atol=1e-4)
