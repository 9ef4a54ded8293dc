"""Property-based tests (hypothesis): expression grammar, block assembler,
unpack round trips."""

import struct

import numpy as np
from hypothesis import given, settings, strategies as st

from srtb_amd.utils.expr import evaluate
from srtb_amd.io import backends as bk
from srtb_amd.io.udp import BlockAssembler
from srtb_amd import ref


# ---- expression grammar vs independently built value ----

@st.composite
def arith_expr(draw, depth=0):
    """Random expression tree; returns (text, value)."""
    if depth > 3 or draw(st.booleans()):
        v = draw(st.integers(min_value=0, max_value=999))
        return str(v), float(v)
    op = draw(st.sampled_from(["+", "-", "*", "paren"]))
    a_t, a_v = draw(arith_expr(depth + 1))
    if op == "paren":
        return f"({a_t})", a_v
    b_t, b_v = draw(arith_expr(depth + 1))
    val = {"+": a_v + b_v, "-": a_v - b_v, "*": a_v * b_v}[op]
    # parenthesize compound operands: the generator does not model the
    # grammar's precedence/associativity, only its value
    return f"({a_t}) {op} ({b_t})", val


@given(arith_expr())
@settings(max_examples=200, deadline=None)
def test_expr_random_trees(tv):
    text, value = tv
    assert abs(evaluate(text) - value) <= 1e-6 * max(1.0, abs(value))


# ---- unpack: packing round trip ----

@given(st.integers(0, 2**32 - 1), st.sampled_from([1, 2, 4]))
@settings(max_examples=200, deadline=None)
def test_unpack_subbyte_roundtrip(word, nbits):
    raw = np.frombuffer(struct.pack("<I", word), dtype=np.uint8)
    out = ref.unpack(raw, nbits).astype(np.uint8)
    # repack MSB-first and compare
    per = 8 // nbits
    repacked = []
    for b in range(4):
        v = 0
        for i in range(per):
            v = (v << nbits) | int(out[b * per + i])
        repacked.append(v)
    assert bytes(repacked) == raw.tobytes()


# ---- block assembler vs a simple dict model ----

@given(st.lists(st.integers(0, 30), min_size=1, max_size=60),
       st.integers(2, 5))
@settings(max_examples=100, deadline=None)
def test_block_assembler_model(counters, ppb):
    payload = bk.FastmbRoach2.packet_payload_size - 8
    asm = BlockAssembler(bk.FastmbRoach2, ppb * payload)
    emitted = []
    for c in counters:
        pkt = struct.pack("<Q", c) + bytes([c % 256]) * payload
        blk = asm.push(pkt)
        if blk is not None:
            emitted.append(blk.copy())
    # invariants (jumps may skip whole blocks, so slot positions are not
    # modeled exactly): every payload segment is either zero-filled (lost)
    # or constant-valued with a value that came from some sent counter
    sent_values = {c % 256 for c in counters}
    for blk in emitted:
        for p in range(len(blk) // payload):
            seg = blk[p * payload:(p + 1) * payload]
            assert (seg == seg[0]).all()
            if seg[0] != 0:
                assert int(seg[0]) in sent_values
    st_ = asm.stats
    assert st_.received <= len(counters)
    assert st_.received + st_.out_of_order + st_.wrong_size <= len(counters)


# ---- nsamps_reserved invariants ----

@given(st.integers(16, 26), st.integers(4, 12),
       st.floats(1.0, 1000.0), st.floats(10.0, 400.0))
@settings(max_examples=100, deadline=None)
def test_nsamps_reserved_invariants(log_n, log_s, dm, bw):
    n, s = 2**log_n, 2**log_s
    r = ref.nsamps_reserved(n, s, 1400.0, -bw, 128e6, -dm)
    assert 0 <= r <= n
    if r > 0:
        assert (n - r) % (2 * s) == 0
