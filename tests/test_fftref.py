"""Validate the FFT index-math oracle against NumPy (CPU)."""

import numpy as np
import pytest

from srtb_amd import fftref


@pytest.mark.parametrize("n", [2, 4, 8, 32, 256, 1024])
@pytest.mark.parametrize("sign", [-1, 1])
def test_fft_small_matches_numpy(n, sign):
    rng = np.random.default_rng(n)
    x = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    out = fftref.fft_small(x, sign)
    expect = np.fft.fft(x) if sign == -1 else np.fft.ifft(x) * n
    np.testing.assert_allclose(out, expect, rtol=1e-4, atol=1e-3)


@pytest.mark.parametrize("l1,l2", [(4, 8), (16, 16), (64, 32), (128, 256)])
@pytest.mark.parametrize("sign", [-1, 1])
def test_fft_four_step(l1, l2, sign):
    n = l1 * l2
    rng = np.random.default_rng(n + sign)
    x = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    out = fftref.fft_four_step(x, l1, l2, sign)
    expect = np.fft.fft(x) if sign == -1 else np.fft.ifft(x) * n
    np.testing.assert_allclose(out, expect, rtol=1e-3, atol=1e-2)


@pytest.mark.parametrize("factors", [[8, 8, 8], [4, 16, 8], [16, 16, 16]])
def test_fft_deep(factors):
    n = int(np.prod(factors))
    rng = np.random.default_rng(n)
    x = (rng.normal(size=n) + 1j * rng.normal(size=n)).astype(np.complex64)
    out = fftref.fft_deep(x, factors, -1)
    np.testing.assert_allclose(out, np.fft.fft(x), rtol=1e-3, atol=1e-2)


@pytest.mark.parametrize("n", [16, 256, 4096])
def test_r2c_post_and_rfft_packed(n):
    rng = np.random.default_rng(n)
    x = rng.normal(size=n).astype(np.float32)
    out = fftref.rfft_packed(x)
    expect = np.fft.rfft(x)[:-1]
    np.testing.assert_allclose(out, expect, rtol=1e-3, atol=1e-3)


def test_rfft_packed_with_composite_core():
    n = 2048
    rng = np.random.default_rng(0)
    x = rng.normal(size=n).astype(np.float32)
    out = fftref.rfft_packed(x, factors=[32, 32])
    expect = np.fft.rfft(x)[:-1]
    np.testing.assert_allclose(out, expect, rtol=1e-3, atol=1e-2)


def test_bit_reverse():
    np.testing.assert_array_equal(fftref.bit_reverse_indices(8),
                                  [0, 4, 2, 6, 1, 5, 3, 7])


@pytest.mark.parametrize("t", list(range(2, 31)))
def test_plan_factors_invariants(t):
    """The factorization policy must always produce a valid plan: product
    = 2**t, at most 4 column factors each <= 64, final a pure power of 4
    in [64, 4096], and >= 2 factors for multi-pass lengths."""
    f = fftref.plan_factors(t)
    assert int(np.prod(f)) == 1 << t
    if t <= 12:
        assert f == [1 << t]
        return
    cols, final = f[:-1], f[-1]
    assert 1 <= len(cols) <= 4
    assert all(2 <= c <= 64 for c in cols)
    assert final in (64, 256, 1024, 4096)
    assert (final.bit_length() - 1) % 2 == 0  # pure 4^k


def test_plan_factors_flagship_shapes():
    """Pin the measured-best plans for the two flagship shapes."""
    assert fftref.plan_factors(29) == [64, 64, 64, 32, 64]  # fwd 2^30 R2C
    assert fftref.plan_factors(18) == [64, 64, 64]           # bwd waterfall
    assert fftref.plan_factors(13) == [32, 256]              # 2-pass small
