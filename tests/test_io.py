import os
import struct

import numpy as np
import pytest

from srtb_amd.io import backends as bk
from srtb_amd.io.file_input import FileBlockReader
from srtb_amd.io.udp import BlockAssembler
from srtb_amd.io import writers as wr


# ---------------- backends ----------------

def test_backend_registry():
    assert bk.get_data_stream_count("simple") == 1
    assert bk.get_data_stream_count("fastmb_roach2") == 1
    assert bk.get_data_stream_count("naocpsr_snap1") == 2
    assert bk.get_data_stream_count("gznupsr_a1") == 2
    assert bk.resolve_alias("naocpsr_roach2") == "fastmb_roach2"
    assert bk.get_backend("naocpsr_roach2") is bk.FastmbRoach2
    with pytest.raises(ValueError):
        bk.get_backend("nope")


def test_roach2_parse_packet():
    payload = bytes(4096)
    pkt = struct.pack("<Q", 0x1122334455667788) + payload
    counter, ts = bk.FastmbRoach2.parse_packet(pkt)
    assert counter == 0x1122334455667788
    assert ts == counter


def test_gznupsr_parse_packet_counter_from_vdif_words():
    words = [0] * 8
    words[6] = 0xDDCCBBAA
    words[7] = 0x00000011
    header = struct.pack("<8I", *words) + bytes(32) + bytes(8192)
    counter, _ = bk.GznupsrA1.parse_packet(header)
    assert counter == 0x11DDCCBBAA


def test_vdif_header_fields():
    w0 = (1 << 31) | (0 << 30) | 12345          # invalid=1, legacy=0, secs
    w1 = (5 << 24) | 678                        # epoch=5, frame count
    w2 = (2 << 29) | (3 << 24) | 1024           # version=2, log2ch=3, length
    w3 = (1 << 31) | (7 << 26) | (9 << 16) | 42  # cplx, bits-1=7, thread, station
    buf = struct.pack("<8I", w0, w1, w2, w3, 0, 0, 0, 0)
    h = bk.VdifHeader.parse(buf)
    assert h.invalid_data == 1 and h.legacy_mode == 0
    assert h.seconds_from_ref_epoch == 12345
    assert h.reference_epoch == 5
    assert h.data_frame_count_in_second == 678
    assert h.vdif_version == 2 and h.log2_channels == 3
    assert h.data_frame_length == 1024
    assert h.data_type == 1 and h.bits_per_sample_minus_1 == 7
    assert h.thread_id == 9 and h.station_id == 42


# ---------------- block assembler ----------------

def mk_packet(counter, payload_byte=None, payload_len=4096):
    payload = bytes([payload_byte if payload_byte is not None
                     else counter % 256]) * payload_len
    return struct.pack("<Q", counter) + payload


def test_assembler_in_order():
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    out = None
    for c in range(5):
        r = asm.push(mk_packet(c))
        if r is not None:
            out = r
    assert out is not None
    for c in range(4):
        assert (out[c * 4096:(c + 1) * 4096] == c % 256).all()
    assert asm.stats.received == 5
    assert asm.stats.lost == 0


def test_assembler_gap_zero_fill():
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    asm.push(mk_packet(0))
    asm.push(mk_packet(1))
    # packet 2 lost
    asm.push(mk_packet(3))
    out = asm.push(mk_packet(4))
    assert out is not None
    assert (out[2 * 4096:3 * 4096] == 0).all()
    assert (out[3 * 4096:4 * 4096] == 3 % 256).all()
    assert asm.stats.lost == 1


def test_assembler_out_of_order_and_wrong_size():
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    asm.push(mk_packet(10))
    assert asm.push(mk_packet(9)) is None  # before begin
    assert asm.stats.out_of_order == 1
    assert asm.push(b"short") is None
    assert asm.stats.wrong_size == 1


def test_assembler_big_jump_counts_lost_blocks():
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    asm.push(mk_packet(0))
    out = asm.push(mk_packet(13))  # jump over 3 whole blocks
    assert out is not None
    # 3 packets of block 0 lost + 2 fully-lost blocks (4 each)
    assert asm.stats.lost == 3 + 2 * 4
    out2 = asm.push(mk_packet(16))
    assert out2 is not None  # block [12..16) completes
    assert (out2[4096:2 * 4096] == 13 % 256).all()


def test_assembler_block_counter_labels_completed_block():
    # each completed block is stamped with ITS OWN begin counter
    # (reference block_first_counter), not the triggering packet's nor the
    # stream's first counter ever
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    for c in range(4):
        assert asm.push(mk_packet(c)) is None
    out = asm.push(mk_packet(4))  # completes block [0..4)
    assert out is not None
    assert asm.last_block_counter == 0
    for c in range(5, 8):
        asm.push(mk_packet(c))
    out = asm.push(mk_packet(8))  # completes block [4..8)
    assert out is not None
    assert asm.last_block_counter == 4
    # loss-crossing completion: jump into a far block — completed block
    # is still labeled with its own begin (8), next begin aligned to 16
    out = asm.push(mk_packet(17))
    assert out is not None
    assert asm.last_block_counter == 8
    out = asm.push(mk_packet(20))  # completes block [16..20)
    assert out is not None
    assert asm.last_block_counter == 16


def test_assembler_duplicates_counted_separately():
    asm = BlockAssembler(bk.FastmbRoach2, 4096 * 4)
    asm.push(mk_packet(0, payload_byte=7))
    asm.push(mk_packet(0, payload_byte=9))  # duplicate, dropped
    assert asm.stats.received == 1
    assert asm.stats.duplicate == 1
    asm.push(mk_packet(1))
    asm.push(mk_packet(2))
    asm.push(mk_packet(3))
    out = asm.push(mk_packet(4))
    assert out is not None
    assert (out[:4096] == 7).all()  # first copy wins
    assert asm.stats.lost == 0  # duplicates must not corrupt loss stats


def test_assembler_simple_headerless_append():
    # 'simple' backend = headerless linear sample stream (reference
    # backend_registry.hpp:36-39): bytes append sequentially, any size
    asm = BlockAssembler(bk.Simple, 100)
    data = bytes(range(250))
    blocks = []
    for off in range(0, 250, 60):  # arbitrary chunking
        blk = asm.push(data[off:off + 60])
        if blk is not None:
            blocks.append((asm.last_block_counter, blk))
    assert len(blocks) == 2
    assert blocks[0][0] == 0 and blocks[1][0] == 100
    assert bytes(blocks[0][1]) == data[:100]
    assert bytes(blocks[1][1]) == data[100:200]


# ---------------- file input ----------------

def test_file_reader_overlap(tmp_path):
    path = tmp_path / "bb.bin"
    data = np.arange(1000, dtype=np.uint8)
    data.tofile(path)
    r = FileBlockReader(str(path), block_samples=256, nbits=8,
                        nsamps_reserved=64)
    blocks = list(r)
    assert r.n_blocks() == len(blocks)
    step = 256 - 64
    for k, (idx, blk) in enumerate(blocks):
        assert idx == k * step
        np.testing.assert_array_equal(
            blk, data[k * step:k * step + 256])
    # consecutive blocks share the 64-byte overlap
    np.testing.assert_array_equal(blocks[0][1][-64:], blocks[1][1][:64])


def test_file_reader_2bit_offset(tmp_path):
    path = tmp_path / "bb2.bin"
    np.arange(300, dtype=np.uint8).tofile(path)
    r = FileBlockReader(str(path), block_samples=512, nbits=2,
                        nsamps_reserved=0, offset_bytes=10)
    blocks = list(r)
    assert len(blocks) == (300 - 10) // 128
    assert blocks[0][1].size == 512 * 2 // 8
    assert blocks[0][1][0] == 10


# ---------------- writers ----------------

def test_writers_file_formats(tmp_path):
    prefix = str(tmp_path / "out_")
    raw = np.arange(64, dtype=np.uint8)
    p1 = wr.write_baseband_bin(prefix, 42, raw)
    assert open(p1, "rb").read() == raw.tobytes()
    wf = (np.arange(12).reshape(3, 4) * (1 + 1j)).astype(np.complex64)
    p2 = wr.write_spectrum_npy(prefix, 42, wf)
    assert p2.endswith("42.0.npy")
    np.testing.assert_array_equal(np.load(p2), wf)
    p2b = wr.write_spectrum_npy(prefix, 42, wf)
    assert p2b.endswith("42.1.npy")  # second pol gets next free index
    ts = np.linspace(0, 1, 10, dtype=np.float32)
    p3 = wr.write_time_series_tim(prefix, 42, 4, ts)
    back = np.frombuffer(open(p3, "rb").read(), dtype=np.float32)
    np.testing.assert_array_equal(back, ts)


def test_signal_write_scheduler_coincidence(tmp_path):
    prefix = str(tmp_path / "s_")
    sched = wr.SignalWriteScheduler(prefix, block_samples=1000,
                                    sample_rate=1e6, real_time=True)
    win = sched.overlap_window_ns
    ts0 = 10_000_000_000
    # negative block first — held back
    blk_neg = wr.BlockProducts(counter=1, timestamp=ts0,
                               raw=np.zeros(4, np.uint8))
    sched.push(blk_neg)
    assert sched.written == []
    # positive block within the window → both written
    blk_pos = wr.BlockProducts(counter=2, timestamp=int(ts0 + win / 2),
                               raw=np.ones(4, np.uint8),
                               time_series=[(1, np.ones(8, np.float32))])
    sched.push(blk_pos)
    assert any("2.bin" in p for p in sched.written)
    assert any("2.1.tim" in p for p in sched.written)
    # held-back negative flushed by coincidence
    sched.push(wr.BlockProducts(counter=3, timestamp=int(ts0 + 10 * win)))
    assert any("1.bin" in p for p in sched.written)


def test_signal_write_scheduler_no_coincidence(tmp_path):
    prefix = str(tmp_path / "n_")
    sched = wr.SignalWriteScheduler(prefix, 1000, 1e6, real_time=True)
    for i in range(5):
        sched.push(wr.BlockProducts(counter=i, timestamp=i * 10**12,
                                    raw=np.zeros(2, np.uint8)))
    assert sched.written == []


def test_signal_write_scheduler_file_mode(tmp_path):
    # file replay: only positive blocks written (no coincidence logic)
    prefix = str(tmp_path / "f_")
    sched = wr.SignalWriteScheduler(prefix, 1000, 1e6, real_time=False)
    sched.push(wr.BlockProducts(counter=0, timestamp=0,
                                raw=np.zeros(2, np.uint8)))
    assert sched.written == []
    sched.push(wr.BlockProducts(counter=1, timestamp=10,
                                time_series=[(2, np.ones(4, np.float32))]))
    assert len(sched.written) == 1


def test_filterbank_header_roundtrip(tmp_path):
    hdr = wr.filterbank_header(fch1=1437.0, foff=-64.0 / 2048, nchans=2048,
                               tsamp=3.2e-5, source_name="J1644-4559")
    assert hdr.startswith(struct.pack("<i", 12) + b"HEADER_START")
    assert hdr.endswith(struct.pack("<i", 10) + b"HEADER_END")
    assert b"src_raj" in hdr and b"nchans" in hdr
    # nchans value encoded little-endian after its key
    i = hdr.index(b"nchans") + 6
    assert struct.unpack_from("<i", hdr, i)[0] == 2048
    data = np.zeros((4, 2048), dtype=np.float32)
    path = str(tmp_path / "t.fil")
    wr.write_filterbank(path, hdr, data)
    assert os.path.getsize(path) == len(hdr) + data.nbytes


def test_to_sigproc_dms():
    assert wr.to_sigproc_dms(12.5) == pytest.approx(123000.0)
    assert wr.to_sigproc_dms(-1.25) == pytest.approx(-11500.0)


def test_signal_write_scheduler_async(tmp_path):
    """async_writes=True posts product writes to a pool; close() flushes —
    files must exist and match the sync-mode output."""
    import numpy as np
    from srtb_amd.io.writers import BlockProducts, SignalWriteScheduler
    raw = np.arange(64, dtype=np.uint8)
    ts = np.linspace(0, 1, 32, dtype=np.float32)
    with SignalWriteScheduler(str(tmp_path) + "/a_", 1 << 20, 1e9,
                              async_writes=True) as sch:
        sch.push(BlockProducts(counter=7, timestamp=0, raw=raw,
                               waterfall=None, time_series=[(1, ts)]))
    files = sorted(p.name for p in tmp_path.iterdir())
    assert files == ["a_7.1.tim", "a_7.bin"]
    got = np.fromfile(tmp_path / "a_7.bin", dtype=np.uint8)
    np.testing.assert_array_equal(got, raw)
