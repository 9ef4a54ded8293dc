"""srtb_amd main application — the reference `simple-radio-telescope-backend`
executable equivalent (reference src/main.cpp:61-333).

Reads a srtb_config.cfg-compatible config (cmd > cfg-file > defaults), sets up
the per-stream pipelines (native GPU engine, or the NumPy path on CPU-only
hosts), consumes baseband from a recorded file (with dedispersion-overlap
seek-back) or UDP, writes detection products (.bin/.npy/.tim) with the
polarization-coincidence scheduler, and optionally dumps live waterfall
frames (headless PPM — the Qt GUI equivalent).

Multi-GPU: launch under torchrun, one rank per GPU; streams are sharded
round-robin (parallel.sharding) and detection stats all-reduced over
RCCL/xGMI.

Usage:
  python -m srtb_amd.main --config_file_name srtb_config.cfg [--key value ...]
  torchrun --nproc-per-node 8 -m srtb_amd.main --config_file_name ...
"""

from __future__ import annotations

import sys
import time

import numpy as np

from . import ref
from .config import Config, parse_args
from .io import backends as bk
from .io.file_input import FileBlockReader
from .io.writers import BlockProducts, SignalWriteScheduler
from .pipeline.cpu import CpuPipeline
from .parallel.sharding import (DetectionAggregator, broadcast_config,
                                init_distributed, shard_streams)

# extra, non-reference options handled by the runner itself
RUNNER_FLAGS = {"--max-blocks", "--device", "--waterfall-ppm", "--gui-port",
                "--gui-linger"}


def write_ppm(path: str, argb: np.ndarray) -> None:
    """Dump an ARGB32 [H][W] image as binary PPM (P6)."""
    h, w = argb.shape
    rgb = np.empty((h, w, 3), dtype=np.uint8)
    rgb[..., 0] = (argb >> 16) & 0xFF
    rgb[..., 1] = (argb >> 8) & 0xFF
    rgb[..., 2] = argb & 0xFF
    with open(path, "wb") as f:
        f.write(b"P6\n%d %d\n255\n" % (w, h))
        f.write(rgb.tobytes())


def _engine_products(eng, cfg: Config, slot: int, res: dict,
                     raw: np.ndarray | None, counter: int) -> BlockProducts:
    """Build the dumpable products of one processed block from a native
    engine slot (detection gate + waterfall/time-series extraction)."""
    # raw baseband is attached even without a detection: the reference's
    # write_signal works always carry baseband_data, so a cross-pol
    # coincidence write can dump the negative stream's baseband too
    products = BlockProducts(counter=counter, timestamp=counter, raw=raw)
    gate = res["zero_count"] < (cfg.signal_detect_channel_threshold *
                                cfg.spectrum_channel_count)
    detected = [(L, c) for L, c in res["counts"] if c > 0]
    if gate and detected:
        products.waterfall = eng.waterfall(slot).cpu().numpy()
        ts = eng.time_series(slot).cpu().numpy()
        for L, _count in detected:
            if L == 1:
                products.time_series.append((1, ts.copy()))
            else:
                products.time_series.append(
                    (L, eng.boxcar_series(slot, L).cpu().numpy()))
    return products


class GpuStreamPipeline:
    """One data stream on one GPU via the native engine."""

    def __init__(self, cfg: Config, nsamps_reserved: int):
        import torch
        from .ops import native
        self.torch = torch
        self.C = native()
        self.cfg = cfg
        ranges = ref.parse_rfi_freq_list(cfg.mitigate_rfi_freq_list)
        nc = cfg.baseband_input_count // 2
        bins = ref.rfi_ranges_to_bins(cfg.baseband_freq_low,
                                      cfg.baseband_bandwidth, nc, ranges)
        self.eng = self.C.PipelineEngine(
            n=cfg.baseband_input_count, nbits=cfg.baseband_input_bits,
            channels=cfg.spectrum_channel_count,
            freq_low=cfg.baseband_freq_low, bandwidth=cfg.baseband_bandwidth,
            sample_rate=cfg.baseband_sample_rate, dm=cfg.dm,
            rfi_threshold=cfg.mitigate_rfi_average_method_threshold,
            sk_threshold=cfg.mitigate_rfi_spectral_kurtosis_threshold,
            snr_threshold=cfg.signal_detect_signal_noise_threshold,
            max_boxcar=cfg.signal_detect_max_boxcar_length,
            nsamps_reserved=nsamps_reserved,
            zap_ranges=[[int(a), int(b)] for a, b in bins],
            use_phase_table=False, enable_rfi_s1=True, enable_sk=True,
            n_slots=2)

    def process_block(self, raw: np.ndarray, counter: int) -> BlockProducts:
        torch = self.torch
        t = torch.from_numpy(np.ascontiguousarray(raw))
        slot = self.eng.submit(t)
        res = self.eng.wait(slot)
        self.last_result = res
        self.last_slot = slot
        return _engine_products(self.eng, self.cfg, slot, res, raw, counter)

    def waterfall_frame(self, width: int, height: int) -> np.ndarray:
        wf = self.eng.waterfall(self.last_slot)
        img = self.C.resample_power(wf, height, width)
        self.C.normalize_by_mean(img)
        pix = self.C.generate_pixmap(img, ref.COLOR_0, ref.COLOR_1,
                                     ref.COLOR_OVERFLOW)
        return pix.cpu().numpy().view(np.uint32)

    def spectrum_lines(self) -> list[np.ndarray]:
        # per-channel mean intensity of the latest waterfall (the
        # reference's live spectrum view, src/spectrum.qml)
        wf = self.eng.waterfall(self.last_slot)
        return [(wf.real ** 2 + wf.imag ** 2).mean(dim=1).cpu().numpy()]


class GpuMultiPolPipeline:
    """One PACKED packet stream fanning out into per-polarization sample
    streams, each through its own engine (reference unpack_pipe.hpp:146-390
    fans one naocpsr_snap1 / gznupsr_a1 stream into data_stream_id works)."""

    def __init__(self, cfg: Config, nsamps_reserved: int, fmt_name: str):
        import torch
        from .ops import native
        from .pipeline.gpu import _make_engine
        self.torch = torch
        self.C = native()
        self.cfg = cfg
        self.fmt = bk.resolve_alias(fmt_name)
        self.n_streams = bk.get_data_stream_count(fmt_name)
        self.engines = [_make_engine(self.C, cfg, nsamps_reserved, nbits=-8)
                        for _ in range(self.n_streams)]

    def _fanout(self, raw_t):
        if self.fmt == "gznupsr_a1":
            return self.C.unpack_gznupsr_a1(raw_t, self.n_streams)
        return self.C.unpack_2pol(raw_t, self.fmt)

    def process_blocks(self, raw: np.ndarray,
                       counter: int) -> list[BlockProducts]:
        torch = self.torch
        raw_t = torch.from_numpy(np.ascontiguousarray(raw)).cuda()
        pols = self._fanout(raw_t)
        slots = [eng.submit_samples(pol)
                 for eng, pol in zip(self.engines, pols)]
        out = []
        results = []
        for eng, slot in zip(self.engines, slots):
            res = eng.wait(slot)
            results.append(res)
            # both pols share the same raw packet block (as the reference's
            # fanned-out works share baseband_data): attach it to each so a
            # coincidence write from either stream dumps it
            out.append(_engine_products(eng, self.cfg, slot, res, raw,
                                        counter))
        self.last_result = results[0]
        self.last_slot = slots[0]
        self.last_slots = slots
        return out

    def _frame(self, eng, slot, width: int, height: int) -> np.ndarray:
        wf = eng.waterfall(slot)
        img = self.C.resample_power(wf, height, width)
        self.C.normalize_by_mean(img)
        pix = self.C.generate_pixmap(img, ref.COLOR_0, ref.COLOR_1,
                                     ref.COLOR_OVERFLOW)
        return pix.cpu().numpy().view(np.uint32)

    def waterfall_frame(self, width: int, height: int) -> np.ndarray:
        return self._frame(self.engines[0], self.last_slot, width, height)

    def waterfall_frames(self, width: int, height: int) -> list[np.ndarray]:
        """One pixmap per data stream (the reference opens one waterfall
        window per stream, src/main.qml:14-28)."""
        return [self._frame(e, s, width, height)
                for e, s in zip(self.engines, self.last_slots)]

    def spectrum_lines(self) -> list[np.ndarray]:
        out = []
        for e, sl in zip(self.engines, self.last_slots):
            wf = e.waterfall(sl)
            out.append((wf.real ** 2 + wf.imag ** 2).mean(dim=1)
                       .cpu().numpy())
        return out


class CpuMultiPolPipeline:
    """CPU/NumPy twin of GpuMultiPolPipeline (oracle + plumbing runs)."""

    def __init__(self, cfg: Config, nsamps_reserved: int, fmt_name: str):
        self.cfg = cfg
        self.fmt = bk.resolve_alias(fmt_name)
        self.n_streams = bk.get_data_stream_count(fmt_name)
        self.pipes = [CpuPipeline(cfg) for _ in range(self.n_streams)]

    def _fanout(self, raw: np.ndarray) -> list[np.ndarray]:
        if self.fmt == "gznupsr_a1":
            return ref.unpack_gznupsr_a1(raw, self.n_streams)
        return list(ref.unpack_naocpsr_snap1(raw))

    def process_blocks(self, raw: np.ndarray,
                       counter: int) -> list[BlockProducts]:
        cfg = self.cfg
        out = []
        self.last_results = []
        for pipe, samples in zip(self.pipes, self._fanout(raw)):
            res = pipe.process_samples(samples)
            self.last_result = res
            self.last_results.append(res)
            products = BlockProducts(counter=counter, timestamp=counter,
                                     raw=raw)
            gate = res["zero_count"] < (cfg.signal_detect_channel_threshold *
                                        cfg.spectrum_channel_count)
            if gate and res["detections"]:
                products.waterfall = res["waterfall"]
                for L, _cnt, series in res["detections"]:
                    products.time_series.append((L, series))
            out.append(products)
        return out

    def waterfall_frame(self, width: int, height: int) -> np.ndarray:
        wf = self.last_result["waterfall"]
        p = np.abs(wf.astype(np.complex64)) ** 2
        img = ref.resample_power_2d(p, height, width)
        img = ref.normalize_by_mean(img)
        return ref.generate_pixmap(img)

    def waterfall_frames(self, width: int, height: int) -> list[np.ndarray]:
        out = []
        for res in self.last_results:
            p = np.abs(res["waterfall"].astype(np.complex64)) ** 2
            img = ref.normalize_by_mean(ref.resample_power_2d(p, height,
                                                              width))
            out.append(ref.generate_pixmap(img))
        return out

    def spectrum_lines(self) -> list[np.ndarray]:
        return [(np.abs(r["waterfall"].astype(np.complex64)) ** 2
                 ).mean(axis=1) for r in self.last_results]


class CpuStreamPipeline:
    """CPU/NumPy path (plumbing runs, no GPU)."""

    def __init__(self, cfg: Config, nsamps_reserved: int):
        self.cfg = cfg
        self.pipe = CpuPipeline(cfg)

    def process_block(self, raw: np.ndarray, counter: int) -> BlockProducts:
        cfg = self.cfg
        res = self.pipe.process_block(raw)
        self.last_result = res
        products = BlockProducts(counter=counter, timestamp=counter, raw=raw)
        gate = res["zero_count"] < (cfg.signal_detect_channel_threshold *
                                    cfg.spectrum_channel_count)
        if gate and res["detections"]:
            products.waterfall = res["waterfall"]
            for L, _cnt, series in res["detections"]:
                products.time_series.append((L, series))
        return products

    def waterfall_frame(self, width: int, height: int) -> np.ndarray:
        wf = self.last_result["waterfall"]
        p = np.abs(wf.astype(np.complex64)) ** 2
        img = ref.resample_power_2d(p, height, width)
        img = ref.normalize_by_mean(img)
        return ref.generate_pixmap(img)

    def spectrum_lines(self) -> list[np.ndarray]:
        wf = self.last_result["waterfall"]
        return [(np.abs(wf.astype(np.complex64)) ** 2).mean(axis=1)]


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    # split runner-only flags from reference-style config options
    max_blocks = None
    device = None
    waterfall_every = 0
    gui_port = 8265
    gui_linger = 0.0
    cfg_argv = []
    i = 0
    while i < len(argv):
        a = argv[i]
        if a.startswith("--max-blocks"):
            max_blocks = int(a.split("=", 1)[1] if "=" in a else argv[i + 1])
            i += 1 if "=" in a else 2
        elif a.startswith("--device"):
            device = a.split("=", 1)[1] if "=" in a else argv[i + 1]
            i += 1 if "=" in a else 2
        elif a.startswith("--waterfall-ppm"):
            waterfall_every = int(a.split("=", 1)[1] if "=" in a else argv[i + 1])
            i += 1 if "=" in a else 2
        elif a.startswith("--gui-port"):
            gui_port = int(a.split("=", 1)[1] if "=" in a else argv[i + 1])
            i += 1 if "=" in a else 2
        elif a.startswith("--gui-linger"):
            gui_linger = float(a.split("=", 1)[1] if "=" in a else argv[i + 1])
            i += 1 if "=" in a else 2
        else:
            cfg_argv.append(a)
            i += 1
    rank, world, local_rank = init_distributed()
    cfg = parse_args(cfg_argv) if rank == 0 else Config()
    cfg = broadcast_config(cfg if rank == 0 else None)

    import importlib
    torch_spec = importlib.util.find_spec("torch")
    use_gpu = False
    if device != "cpu" and torch_spec is not None:
        import torch
        use_gpu = torch.cuda.is_available()

    reserved = ref.nsamps_reserved(
        cfg.baseband_input_count, cfg.spectrum_channel_count,
        cfg.baseband_freq_low, cfg.baseband_bandwidth,
        cfg.baseband_sample_rate, cfg.dm, cfg.baseband_reserve_sample)

    # polarization fan-out: formats with >1 data stream run one packed
    # packet stream through per-pol pipelines + cross-pol coincidence writes
    n_streams = bk.get_data_stream_count(cfg.baseband_format_type)

    def make(cfg_, reserved_):
        if n_streams > 1:
            cls = GpuMultiPolPipeline if use_gpu else CpuMultiPolPipeline
            return cls(cfg_, reserved_, cfg.baseband_format_type)
        return (GpuStreamPipeline if use_gpu else CpuStreamPipeline)(
            cfg_, reserved_)

    def process(pipe, raw, counter) -> list[BlockProducts]:
        if hasattr(pipe, "process_blocks"):
            return pipe.process_blocks(raw, counter)
        return [pipe.process_block(raw, counter)]

    # live waterfall GUI (reference Qt windows → built-in browser viewer);
    # only rank 0 serves when running under torchrun
    gui = None
    if cfg.gui_enable and rank == 0:
        from .gui import WaterfallServer
        gui = WaterfallServer(port=gui_port).start()
        print(f"[srtb_amd] live waterfall: http://127.0.0.1:{gui.port}/")

    def gui_update(pipe, blocks_done, written):
        if gui is None:
            return
        try:
            if hasattr(pipe, "waterfall_frames"):
                frames = pipe.waterfall_frames(cfg.gui_pixmap_width,
                                               cfg.gui_pixmap_height)
            else:
                frames = [pipe.waterfall_frame(cfg.gui_pixmap_width,
                                               cfg.gui_pixmap_height)]
        except Exception:
            return  # no block processed yet
        for sid, frame in enumerate(frames):
            gui.push_frame(sid, frame)
        try:
            for sid, line in enumerate(pipe.spectrum_lines()):
                gui.push_spectrum(sid, line)
        except Exception:
            pass
        gui.update_status(blocks=blocks_done, written=written)

    agg = DetectionAggregator()
    writer = SignalWriteScheduler(
        cfg.baseband_output_file_prefix, cfg.baseband_input_count,
        cfg.baseband_sample_rate, real_time=(cfg.input_file_path == ""),
        async_writes=True)

    t0 = time.time()
    n_blocks = 0
    # baseband_write_all: record every block (minus the overlap tail) into
    # one file per stream (reference write_file_pipe.hpp:41-94)
    write_all_f = None
    if cfg.baseband_write_all:
        write_all_f = open(cfg.baseband_output_file_prefix +
                           f"all_r{rank}.bin", "ab")
        reserved_bytes = reserved * abs(cfg.baseband_input_bits) // 8

    if cfg.input_file_path:
        # file replay: one packet/sample stream (multi-pol formats carry
        # n_streams interleaved sample streams per block)
        pipe = make(cfg, reserved)
        reader = FileBlockReader(cfg.input_file_path,
                                 cfg.baseband_input_count * n_streams,
                                 cfg.baseband_input_bits, reserved * n_streams,
                                 cfg.input_file_offset_bytes)
        for sample_index, raw in reader:
            for products in process(pipe, raw, sample_index):
                writer.push(products)
            if write_all_f is not None:
                end = raw.size - reserved_bytes if reserved_bytes else raw.size
                write_all_f.write(raw[:end].tobytes())
            res = getattr(pipe, "last_result", None)
            if isinstance(res, dict) and "zero_count" in res:
                zc = res["zero_count"]
                counts = (res.get("counts")
                          or [(L, c) for L, c, _ in res.get("detections", [])])
            else:
                zc, counts = 0, []
            agg.update(int(zc), [(int(a), int(b)) for a, b in counts])
            n_blocks += 1
            gui_update(pipe, n_blocks, len(writer.written))
            if waterfall_every and n_blocks % waterfall_every == 0:
                frame = pipe.waterfall_frame(cfg.gui_pixmap_width,
                                             cfg.gui_pixmap_height)
                write_ppm(f"{cfg.baseband_output_file_prefix}"
                          f"waterfall_r{rank}_{n_blocks}.ppm", frame)
            if max_blocks is not None and n_blocks >= max_blocks:
                break
    else:
        # UDP ingest: endpoints sharded across ranks; each rank runs one
        # receiver thread per assigned endpoint (the reference spawns N udp
        # receiver pipes, main.cpp:230-272), feeding a shared work queue
        import queue as _queue
        import threading
        from .io.udp import BlockAssembler, UdpPacketProvider, run_receiver
        backend = bk.get_backend(cfg.baseband_format_type)
        n_endpoints = len(cfg.udp_receiver_address)
        my = shard_streams(n_endpoints, world, rank)
        if not my:
            print(f"rank {rank}: no UDP endpoints assigned")
            return 0
        bits = abs(cfg.baseband_input_bits)
        block_bytes = cfg.baseband_input_count * bits // 8 * \
            bk.get_data_stream_count(cfg.baseband_format_type)
        pipes = {ep: make(cfg, reserved) for ep in my}
        q: "_queue.Queue" = _queue.Queue(maxsize=2 * len(my))
        state = {"n": 0, "stop": False}

        def stop():
            return state["stop"] or (max_blocks is not None and
                                     state["n"] >= max_blocks)

        def receiver(ep):
            assembler = BlockAssembler(backend, block_bytes)
            provider = UdpPacketProvider(cfg.udp_receiver_address[ep],
                                         cfg.udp_receiver_port[ep])
            try:
                run_receiver(provider, assembler,
                             lambda blk, ts: q.put((ep, blk, ts)), stop)
            finally:
                provider.close()

        threads = [threading.Thread(target=receiver, args=(ep,), daemon=True)
                   for ep in my]
        for t in threads:
            t.start()
        while not stop():
            try:
                ep, blk, ts = q.get(timeout=0.2)
            except _queue.Empty:
                continue
            for products in process(pipes[ep], blk, ts):
                writer.push(products)
            state["n"] += 1
            gui_update(pipes[ep], state["n"], len(writer.written))
        state["stop"] = True
        n_blocks = state["n"]

    if write_all_f is not None:
        write_all_f.close()
    elapsed = time.time() - t0
    if gui is not None:
        # keep serving the last frames briefly (the reference GUI keeps its
        # windows open after the pipeline drains)
        if gui_linger > 0:
            time.sleep(gui_linger)
        gui.stop()
    writer.close()  # flush the async write pool before the summary
    stats = agg.reduce()
    if rank == 0:
        sps = stats.blocks * cfg.baseband_input_count / max(elapsed, 1e-9)
        print(f"[srtb_amd] blocks={stats.blocks} detections={stats.detections} "
              f"signal_counts={stats.signal_counts} "
              f"zapped_channels={stats.zapped_channels} "
              f"written={len(writer.written)} "
              f"throughput={sps / 1e6:.1f} Msamples/s "
              f"real_time_ratio={sps / cfg.baseband_sample_rate:.2f}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
