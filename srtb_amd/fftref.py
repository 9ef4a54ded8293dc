"""Reference implementation of the hand-written FFT path (index-math oracle).

This file defines, in NumPy, EXACTLY the algorithm the HIP kernels in
csrc/kernels/fft.hip implement:

 1. ``fft_small``  — power-of-two DIT FFT: bit-reversed gather at load time,
    then in-place butterflies (this is what one workgroup does in LDS).
 2. ``fft_four_step`` — composite length L = L1 * L2 (L1, L2 small):
    pass 1: for each column n2 of the [L1][L2] row-major view, an L1-point
            FFT over stride-L2 elements, multiplied by the inter-pass
            twiddle w(sign * n2 * k1 / L), stored back in place;
    pass 2: for each row k1, an L2-point FFT over the contiguous row,
            scattered to out[k1 + L1 * k2] — the "transpose" is folded into
            pass 2's store addressing, so there is no separate transpose
            kernel (rocFFT's plan for these shapes runs 2 transposes).
 3. ``fft_six_step_deep`` — length L1 * L2 * L3 via recursion (2^29 forward).
 4. ``r2c_post`` — real-input FFT via the packed-complex trick: N reals are
    viewed as N/2 complex, C2C-transformed, then split into the true R2C
    spectrum (reference fft/fft_1d_r2c_post_process.hpp:33-82 capability).

Also serves as the "naive FFT" debug fallback of the reference
(fft/naive_fft.hpp K8-K11 in SURVEY.md §2b).

Conventions follow cuFFT/hipFFT: forward sign = -1, backward sign = +1, both
unnormalized.
"""

from __future__ import annotations

import numpy as np


def bit_reverse_indices(n: int) -> np.ndarray:
    t = n.bit_length() - 1
    idx = np.arange(n)
    rev = np.zeros(n, dtype=np.int64)
    for b in range(t):
        rev |= ((idx >> b) & 1) << (t - 1 - b)
    return rev


def twiddle_table(n: int, sign: int) -> np.ndarray:
    """tw[j] = exp(sign * 2πi * j / n), j in [0, n/2) — all stages index into
    this single table: stage with butterfly length ``len`` uses
    tw[j * (n // len)] for j in [0, len/2)."""
    j = np.arange(n // 2)
    return np.exp(sign * 2j * np.pi * j / n).astype(np.complex64)


def fft_small(x: np.ndarray, sign: int, dtype=np.complex64) -> np.ndarray:
    """Power-of-two DIT FFT: bit-reversed load + in-place butterflies.

    Mirrors the workgroup-level LDS algorithm of the HIP kernel.
    """
    n = x.size
    assert n & (n - 1) == 0
    tw = twiddle_table(n, sign).astype(np.complex128)
    y = np.asarray(x, dtype=np.complex128)[bit_reverse_indices(n)].copy()
    length = 2
    while length <= n:
        half = length // 2
        tstep = n // length
        # butterflies: for each group g, element j
        for b in range(n // 2):
            g, j = divmod(b, half)
            i0 = g * length + j
            i1 = i0 + half
            w = tw[j * tstep]
            a, c = y[i0], y[i1] * w
            y[i0] = a + c
            y[i1] = a - c
        length *= 2
    return y.astype(dtype)


def twiddle_table_full(n: int, sign: int) -> np.ndarray:
    """Full-circle table tw[j] = exp(sign*2πi*j/n), j in [0, n) — the radix-4
    stages need indices up to 3n/4."""
    j = np.arange(n)
    return np.exp(sign * 2j * np.pi * j / n)


def fft_small_r4(x: np.ndarray, sign: int, dtype=np.complex64) -> np.ndarray:
    """Mixed radix-4/radix-2 Stockham (ping-pong, auto-sort) — EXACTLY the
    HIP kernel's stage structure: radix-4 stages while n_cur % 4 == 0, one
    final radix-2 stage when log2(n) is odd.

    Stage (radix 4), n_cur, s: m = n_cur/4; for p in [0,m), q in [0,s):
      a,b,c,d = x[q+s(p+km)] k=0..3;  si = sign*1j
      u0 = a+b+c+d; u1 = a+si*b-c-si*d; u2 = a-b+c-d; u3 = a-si*b-c+si*d
      y[q+s(4p+j)] = u_j * w^(j*p),  w = exp(sign*2πi/n_cur)
    then n_cur /= 4, s *= 4.
    """
    n = x.size
    assert n & (n - 1) == 0
    tw = twiddle_table_full(n, sign)
    X = np.asarray(x, dtype=np.complex128).copy()
    Y = np.empty_like(X)
    si = sign * 1j
    n_cur, s = n, 1
    while n_cur % 4 == 0 and n_cur > 1:
        m = n_cur // 4
        tstep = n // n_cur
        for p in range(m):
            w1 = tw[p * tstep]
            w2 = tw[2 * p * tstep]
            w3 = tw[3 * p * tstep]
            for q in range(s):
                a = X[q + s * p]
                b = X[q + s * (p + m)]
                c = X[q + s * (p + 2 * m)]
                d = X[q + s * (p + 3 * m)]
                u0 = a + b + c + d
                u1 = a + si * b - c - si * d
                u2 = a - b + c - d
                u3 = a - si * b - c + si * d
                Y[q + s * (4 * p + 0)] = u0
                Y[q + s * (4 * p + 1)] = u1 * w1
                Y[q + s * (4 * p + 2)] = u2 * w2
                Y[q + s * (4 * p + 3)] = u3 * w3
        X, Y = Y, X
        n_cur //= 4
        s *= 4
    if n_cur == 2:
        m = 1
        tstep = n // 2
        for q in range(s):
            a = X[q]
            b = X[q + s]
            Y[q] = a + b
            Y[q + s] = a - b
        X, Y = Y, X
    return X.astype(dtype)


def digit_reverse4(n: int) -> np.ndarray:
    """Base-4 digit reversal permutation for n = 4^t."""
    t = 0
    while 4**t < n:
        t += 1
    assert 4**t == n
    idx = np.arange(n)
    out = np.zeros(n, dtype=np.int64)
    for d in range(t):
        out = out * 4 + (idx >> (2 * d)) % 4
    return out


def fft_small_dit_r4(x: np.ndarray, sign: int, dtype=np.complex64) -> np.ndarray:
    """Pure radix-4 in-place DIT FFT (n = 4^t): digit-reversed load, then
    in-place butterflies — the register-resident column kernel's algorithm
    (all data stays in one array; no ping-pong).

    Stage with output sub-size len (4, 16, ..., n), quarter q = len/4:
      for each group g (step len), j in [0, q):
        w = exp(sign*2πi*j/len);  b*=w; c*=w²; d*=w³
        t0 = a + c; t1 = a - c; t2 = b + d; t3 = si*(b - d)
        v[g+j]      = t0 + t2
        v[g+j+q]    = t1 + t3
        v[g+j+2q]   = t0 - t2
        v[g+j+3q]   = t1 - t3
    """
    n = x.size
    v = np.asarray(x, dtype=np.complex128)[digit_reverse4(n)].copy()
    si = sign * 1j
    ln = 4
    while ln <= n:
        q = ln // 4
        for g in range(0, n, ln):
            for j in range(q):
                w1 = np.exp(sign * 2j * np.pi * j / ln)
                w2 = w1 * w1
                w3 = w2 * w1
                a = v[g + j]
                b = v[g + j + q] * w1
                c = v[g + j + 2 * q] * w2
                d = v[g + j + 3 * q] * w3
                t0 = a + c
                t1 = a - c
                t2 = b + d
                t3 = si * (b - d)
                v[g + j] = t0 + t2
                v[g + j + q] = t1 + t3
                v[g + j + 2 * q] = t0 - t2
                v[g + j + 3 * q] = t1 - t3
        ln *= 4
    return v.astype(dtype)


def fft_four_step(x: np.ndarray, l1: int, l2: int, sign: int,
                  fft1=None, fft2=None) -> np.ndarray:
    """Composite FFT of length l1*l2 per the docstring above."""
    n = l1 * l2
    assert x.size == n
    fft1 = fft1 or (lambda v: fft_small(v, sign, np.complex128))
    fft2 = fft2 or (lambda v: fft_small(v, sign, np.complex128))
    m = np.asarray(x, dtype=np.complex128).reshape(l1, l2).copy()
    # pass 1: columns (stride l2), then inter-pass twiddle
    for n2 in range(l2):
        col = fft1(m[:, n2])
        k1 = np.arange(l1)
        col = col * np.exp(sign * 2j * np.pi * (k1 * n2) / n)
        m[:, n2] = col
    # pass 2: rows (contiguous), scatter to out[k1 + l1*k2]
    out = np.empty(n, dtype=np.complex128)
    for k1 in range(l1):
        row = fft2(m[k1, :])
        out[k1 + l1 * np.arange(l2)] = row
    return out.astype(np.complex64)


def fft_deep(x: np.ndarray, factors: list[int], sign: int) -> np.ndarray:
    """Arbitrary-depth composite: factors [f0, f1, ..., fk]; recursion
    fft(len=f0 * rest) = four_step with l1=f0, l2=rest (pass-1 FFTs of f0,
    pass-2 = recursive composite of the rest)."""
    if len(factors) == 1:
        return fft_small(x, sign, np.complex128).astype(np.complex64)
    l1 = factors[0]
    l2 = int(np.prod(factors[1:]))
    return fft_four_step(
        x, l1, l2, sign,
        fft1=lambda v: fft_small(v, sign, np.complex128),
        fft2=lambda v: fft_deep(v, factors[1:], sign).astype(np.complex128))


def r2c_post(z: np.ndarray, sign: int = -1) -> np.ndarray:
    """Packed-real R2C: given Z = C2C_fft(x_even + i*x_odd) of length M=N/2,
    recover the true R2C spectrum X[0..M-1] (Nyquist dropped, matching the
    pipeline's spectrum count Nc = N/2).

    X[k] = (Z[k] + conj(Z[M-k]))/2 + w(k) * (Z[k] - conj(Z[M-k]))/(2i),
    w(k) = exp(sign*2πi*k/N); Z[M] := Z[0].
    """
    m = z.size
    zf = np.asarray(z, dtype=np.complex128)
    zk = zf
    zmk = np.conj(np.roll(zf[::-1], 1))  # conj(Z[M-k]), k=0..M-1
    k = np.arange(m)
    w = np.exp(sign * 2j * np.pi * k / (2 * m))
    even = 0.5 * (zk + zmk)
    odd = -0.5j * (zk - zmk)
    return (even + w * odd).astype(np.complex64)


def rfft_packed(x: np.ndarray, factors: list[int] | None = None) -> np.ndarray:
    """Full real-input forward FFT via the packed trick; returns Nc bins."""
    x = np.asarray(x, dtype=np.float64)
    n = x.size
    z = x[0::2] + 1j * x[1::2]
    if factors is None:
        zf = np.fft.fft(z)
    else:
        zf = fft_deep(z.astype(np.complex64), factors, -1).astype(np.complex128)
    return r2c_post(zf, -1)


def plan_factors(t: int, maxcol_log2: int = 6, final_log2: int | None = None) -> list[int]:
    """Mirror of the native planner's factorization policy
    (csrc/fft/native_fft.h plan()): the column factors + final DIF length
    chosen for a 2**t transform.  Kept as the executable SPEC of the
    policy — the C++ is the implementation of record.

    t <= 12 -> [2**t] (single LDS Stockham pass).
    Else: final DIF length 64 (pure 4^k; 256 at t=13), and the residual
    bits go to register-column factors: greedy 64s for rest >= 18 (the
    N=64 lane-pair kernel is the fastest pass; fft_factor_sweep.py),
    balanced <=64 columns otherwise.
    """
    if t <= 12:
        return [1 << t]
    if final_log2 is None:
        # default final DIF length 64 for t >= 14 (r02 dif64 sweep:
        # shorter final pass at 8+ WG/CU wins); 256 at t=13 saves a pass
        final_log2 = 6 if t >= 14 else 8
    if final_log2 % 2:
        final_log2 += 1
    final_log2 = min(max(final_log2, 6), 12)
    if final_log2 >= t:
        final_log2 = t - 1 if t % 2 else t - 2
    rest = t - final_log2
    f: list[int] = []
    if 18 <= rest <= 24:
        left = rest
        while left >= 6 and len(f) < 4:
            f.append(64)
            left -= 6
        if left:
            f.append(1 << left)
    else:
        ncols = min((rest + maxcol_log2 - 1) // maxcol_log2, 4)
        base, extra = divmod(rest, ncols)
        for i in range(ncols):
            b = base + (1 if i < extra else 0)
            assert b <= 6, "factor too large"
            f.append(1 << b)
    f.append(1 << final_log2)
    return f
