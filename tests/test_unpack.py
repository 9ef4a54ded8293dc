import numpy as np
import pytest

from srtb_amd import ref


def test_unpack_1bit_golden():
    # byte 0b01100011 → bits MSB-first
    out = ref.unpack(np.array([0b01100011], dtype=np.uint8), 1)
    assert out.tolist() == [0, 1, 1, 0, 0, 0, 1, 1]


def test_unpack_2bit_golden():
    out = ref.unpack(np.array([0b01100011], dtype=np.uint8), 2)
    assert out.tolist() == [0b01, 0b10, 0b00, 0b11]


def test_unpack_4bit_golden():
    out = ref.unpack(np.array([0b01100011], dtype=np.uint8), 4)
    assert out.tolist() == [0b0110, 0b0011]


def test_unpack_8bit_signed_unsigned():
    raw = np.array([0, 127, 128, 255], dtype=np.uint8)
    assert ref.unpack(raw, 8).tolist() == [0, 127, 128, 255]
    assert ref.unpack(raw, -8).tolist() == [0, 127, -128, -1]


def test_unpack_16bit():
    raw = np.array([1, 0, 0xFF, 0xFF], dtype=np.uint8)  # little endian
    assert ref.unpack(raw, 16).tolist() == [1, 65535]
    assert ref.unpack(raw, -16).tolist() == [1, -1]


def test_unpack_with_window():
    raw = np.array([0b11000000], dtype=np.uint8)
    w = np.array([0.5, 2.0, 1.0, 1.0], dtype=np.float32)
    out = ref.unpack(raw, 2, window=w)
    assert out.tolist() == [1.5, 0.0, 0.0, 0.0]


def test_unpack_sizes():
    rng = np.random.default_rng(0)
    raw = rng.integers(0, 256, 1024, dtype=np.uint8)
    for bits, factor in [(1, 8), (2, 4), (4, 2), (8, 1), (-8, 1)]:
        out = ref.unpack(raw, bits)
        assert out.size == 1024 * factor
        assert out.dtype == np.float32


def test_unpack_interleaved_2pol():
    raw = np.array([1, 2, 3, 4, 5, 6], dtype=np.uint8)
    p0, p1 = ref.unpack_interleaved_2pol(raw)
    assert p0.tolist() == [1, 3, 5]
    assert p1.tolist() == [2, 4, 6]


def test_unpack_naocpsr_snap1():
    # layout: [s0p0, s1p0, s0p1, s1p1] per 4-byte group
    raw = np.array([1, 2, 11, 12, 3, 4, 13, 14], dtype=np.uint8)
    p0, p1 = ref.unpack_naocpsr_snap1(raw)
    assert p0.tolist() == [1, 2, 3, 4]
    assert p1.tolist() == [11, 12, 13, 14]


def test_unpack_gznupsr_a1_2stream():
    # 4-byte words alternating between 2 streams, int8 values
    raw = np.array([1, 2, 3, 4, 201, 202, 203, 204,
                    5, 6, 7, 8, 205, 206, 207, 208], dtype=np.uint8)
    s = ref.unpack_gznupsr_a1(raw, n_streams=2)
    assert s[0].tolist() == [1, 2, 3, 4, 5, 6, 7, 8]
    assert s[1].tolist() == [201 - 256, 202 - 256, 203 - 256, 204 - 256,
                             205 - 256, 206 - 256, 207 - 256, 208 - 256]


def test_unpack_gznupsr_a1_4stream_offset_binary():
    # 4-stream variant XORs 0x80 (offset binary → int8)
    raw = np.zeros(16, dtype=np.uint8)
    raw[0:4] = [0x80, 0x81, 0x7F, 0x00]  # stream 0
    s = ref.unpack_gznupsr_a1(raw, n_streams=4)
    assert s[0].tolist() == [0, 1, -1, -128]


def test_window_hamming_matches_numpy():
    # reference tests hamming against numpy.hamming(16)
    # (tests/test-fft_window.cpp:23-60); numpy uses 0.54/0.46 while the
    # reference uses exact 25/46, 21/46 — compare at matching coefficients
    n = 16
    w = ref.window_coefficients("hamming", n, dtype=np.float64)
    x = np.arange(n) / (n - 1)
    expect = 25 / 46 - 21 / 46 * np.cos(2 * np.pi * x)
    np.testing.assert_allclose(w, expect, rtol=1e-12)
    assert abs(w[0] - (25 / 46 - 21 / 46)) < 1e-12


def test_window_hann():
    w = ref.window_coefficients("hann", 17, dtype=np.float64)
    assert abs(w[0]) < 1e-12
    assert abs(w[8] - 1.0) < 1e-12
    assert abs(w[16]) < 1e-12


def test_window_rectangle():
    assert (ref.window_coefficients("rectangle", 8) == 1).all()
