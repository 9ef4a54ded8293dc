#!/usr/bin/env python3
"""NumPy lane-level simulation of the wave-local FFT kernel (k_fft_wave).

n = 64 * E  (E in {4, 8, 16}): per-lane E-point FFT (DIT over the stride-64
dimension) -> inter-step twiddle W_n^(lane*k2) -> 64-point DIF FFT across
lanes via shfl_xor butterflies (bit-reversed lane output) -> LDS transpose
store.  Validates every index/twiddle choice before the HIP kernel.
"""

import numpy as np


def bitrev(x, bits):
    r = 0
    for _ in range(bits):
        r = (r << 1) | (x & 1)
        x >>= 1
    return r


def wave_fft(x, sign=-1):
    n = x.size
    E = n // 64
    W = np.exp(sign * 2j * np.pi / n)

    # registers: v[lane][k2]; load v[lane][i2] = x[lane + 64*i2]
    v = np.empty((64, E), dtype=complex)
    for lane in range(64):
        # step 1: E-point FFT over i2 (numpy does the per-lane FFT)
        col = x[lane::64]
        if sign < 0:
            v[lane] = np.fft.fft(col)
        else:
            v[lane] = np.fft.ifft(col) * E
    # step 2: twiddle W_n^(lane * k2)
    for lane in range(64):
        for k2 in range(E):
            v[lane][k2] *= W ** (lane * k2)
    # step 3: 64-point DIF across lanes (shfl_xor butterflies), for each k2
    # independently; output lane l holds k1 = bitrev6(l)
    W64 = np.exp(sign * 2j * np.pi / 64)
    M = 32
    while M >= 1:
        nv = v.copy()
        for lane in range(64):
            partner = lane ^ M
            if lane & M == 0:
                nv[lane] = v[lane] + v[partner]
            else:
                j = lane & (M - 1)
                # DIF twiddle of the CURRENT sub-fft length L = 2M:
                # W_L^j = W64^(j * 64/(2M)) = W64^(j * 32/M)
                tw = W64 ** (j * (32 // M))
                nv[lane] = (v[partner] - v[lane]) * tw
        v = nv
        M //= 2
    # store: lane l, register k2 -> X[k2 + E * bitrev6(l)]
    X = np.empty(n, dtype=complex)
    for lane in range(64):
        k1 = bitrev(lane, 6)
        for k2 in range(E):
            X[k2 + E * k1] = v[lane][k2]
    return X


def main():
    rng = np.random.default_rng(0)
    for E in (4, 8, 16):
        n = 64 * E
        x = rng.normal(size=n) + 1j * rng.normal(size=n)
        for sign in (-1, 1):
            got = wave_fft(x, sign)
            ref = np.fft.fft(x) if sign < 0 else np.fft.ifft(x) * n
            err = np.abs(got - ref).max() / np.abs(ref).max()
            print(f"n={n} sign={sign:+d} err={err:.2e}")
            assert err < 1e-10, "index math wrong"
    print("wave FFT index math OK")


if __name__ == "__main__":
    main()
