"""Live waterfall GUI — a built-in browser viewer.

Capability parity with the reference's Qt5/QML live spectrum windows
(reference gui/gui.hpp:34-67, gui/spectrum_image_provider.hpp:331-419,
src/main.qml:14-28): one waterfall view per data stream, updated as blocks
flow through the pipeline.  Qt is not available in this image, so the GUI is
a dependency-free HTTP server: the pipeline pushes ARGB32 pixmaps (the same
`generate_pixmap` chain the reference feeds its QImage) and any browser
renders them live.

Design notes:
- The server holds only the LATEST frame per stream; a slow viewer simply
  skips frames — the same drop-under-load semantics as the reference's
  `loose_queue_out_functor` GUI branch (framework/pipe_io.hpp:79-94).
- Frames are served as 32-bit BMP (ARGB32 little-endian *is* BMP's BGRA
  memory layout, so encoding is a 54-byte header + the raw rows) — no image
  libraries needed, every browser displays it.
- `/` serves a self-refreshing page, `/status.json` the run counters,
  `/frame<i>.bmp` the latest pixmap of stream i.
"""

from __future__ import annotations

import json
import struct
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import numpy as np

_PAGE = """<!DOCTYPE html>
<html><head><title>srtb_amd live waterfall</title>
<style>
 body {{ background:#101018; color:#d0d0e0; font-family:monospace; }}
 .wf {{ border:1px solid #445; margin:4px; image-rendering:pixelated; }}
 .sp {{ border:1px solid #445; margin:4px; display:block; }}
 #status {{ margin:8px; white-space:pre; }}
</style></head>
<body>
<h3>srtb_amd &mdash; live waterfall + spectrum</h3>
<div id="imgs"></div>
<div id="status"></div>
<script>
const REFRESH_MS = {refresh_ms};
let streams = [];
function drawSpectrum(canvas, values) {{
  const ctx = canvas.getContext('2d');
  const w = canvas.width, h = canvas.height;
  ctx.fillStyle = '#101018'; ctx.fillRect(0, 0, w, h);
  if (!values || !values.length) return;
  let lo = Math.min(...values), hi = Math.max(...values);
  if (hi <= lo) hi = lo + 1;
  ctx.strokeStyle = '#6fc36f'; ctx.beginPath();
  for (let i = 0; i < values.length; i++) {{
    const x = i * (w - 1) / (values.length - 1 || 1);
    const y = h - 1 - (values[i] - lo) / (hi - lo) * (h - 2);
    if (i === 0) ctx.moveTo(x, y); else ctx.lineTo(x, y);
  }}
  ctx.stroke();
}}
async function poll() {{
  try {{
    const r = await fetch('/status.json'); const st = await r.json();
    document.getElementById('status').textContent =
      JSON.stringify(st, null, 1);
    if (st.streams.length !== streams.length) {{
      streams = st.streams;
      const div = document.getElementById('imgs'); div.innerHTML = '';
      for (const s of streams) {{
        const img = document.createElement('img');
        img.id = 'wf' + s; img.className = 'wf';
        div.appendChild(img);
        const cv = document.createElement('canvas');
        cv.id = 'sp' + s; cv.className = 'sp';
        cv.width = 640; cv.height = 120;
        div.appendChild(cv);
      }}
    }}
    for (const s of streams) {{
      document.getElementById('wf' + s).src =
        '/frame' + s + '.bmp?t=' + Date.now();
      try {{
        const sr = await fetch('/spectrum' + s + '.json');
        if (sr.ok) {{
          const sj = await sr.json();
          drawSpectrum(document.getElementById('sp' + s), sj.values);
        }}
      }} catch (e) {{}}
    }}
  }} catch (e) {{ /* server gone */ }}
}}
setInterval(poll, REFRESH_MS); poll();
</script>
</body></html>
"""


def encode_bmp(argb: np.ndarray) -> bytes:
    """Encode an ARGB32 [H][W] uint32 array as a 32-bpp BMP.

    ARGB32 words little-endian are B,G,R,A bytes in memory — exactly BMP's
    pixel layout; rows are stored bottom-up.
    """
    h, w = argb.shape
    rows = np.ascontiguousarray(argb[::-1].astype("<u4")).tobytes()
    header = struct.pack(
        "<2sIHHI" "IiiHHIIiiII",
        b"BM", 54 + len(rows), 0, 0, 54,
        40, w, h, 1, 32, 0, len(rows), 2835, 2835, 0, 0)
    return header + rows


class _Handler(BaseHTTPRequestHandler):
    server_version = "srtb-gui/1.0"

    def log_message(self, *a):  # quiet
        pass

    def do_GET(self):  # noqa: N802 (stdlib API)
        srv: "WaterfallServer" = self.server.owner  # type: ignore[attr-defined]
        path = self.path.split("?", 1)[0]
        if path == "/" or path == "/index.html":
            body = _PAGE.format(refresh_ms=srv.refresh_ms).encode()
            self._reply(200, "text/html", body)
        elif path == "/status.json":
            self._reply(200, "application/json",
                        json.dumps(srv.status()).encode())
        elif path.startswith("/spectrum") and path.endswith(".json"):
            try:
                sid = int(path[len("/spectrum"):-len(".json")])
            except ValueError:
                return self._reply(404, "text/plain", b"bad stream")
            spec = srv.get_spectrum_json(sid)
            if spec is None:
                return self._reply(404, "text/plain", b"no spectrum yet")
            self._reply(200, "application/json", spec)
        elif path.startswith("/frame") and path.endswith(".bmp"):
            try:
                sid = int(path[len("/frame"):-len(".bmp")])
            except ValueError:
                return self._reply(404, "text/plain", b"bad stream")
            frame = srv.get_frame_bmp(sid)
            if frame is None:
                return self._reply(404, "text/plain", b"no frame yet")
            self._reply(200, "image/bmp", frame)
        else:
            self._reply(404, "text/plain", b"not found")

    def _reply(self, code: int, ctype: str, body: bytes):
        self.send_response(code)
        self.send_header("Content-Type", ctype)
        self.send_header("Content-Length", str(len(body)))
        self.send_header("Cache-Control", "no-store")
        self.end_headers()
        self.wfile.write(body)


class WaterfallServer:
    """Serve live waterfall frames over HTTP.

    Usage:
        gui = WaterfallServer(port=8265).start()
        gui.push_frame(0, argb)            # per block, per stream
        gui.update_status(blocks=n, ...)   # run counters
        gui.stop()
    """

    def __init__(self, host: str = "0.0.0.0", port: int = 8265,
                 refresh_ms: int = 200):
        self.host = host
        self.port = port
        self.refresh_ms = refresh_ms
        self._frames: dict[int, bytes] = {}
        self._spectra: dict[int, bytes] = {}
        self._status: dict = {}
        self._lock = threading.Lock()
        self._httpd: ThreadingHTTPServer | None = None
        self._thread: threading.Thread | None = None

    def start(self) -> "WaterfallServer":
        self._httpd = ThreadingHTTPServer((self.host, self.port), _Handler)
        self._httpd.owner = self  # type: ignore[attr-defined]
        self.port = self._httpd.server_address[1]  # resolve port 0
        self._thread = threading.Thread(target=self._httpd.serve_forever,
                                        name="srtb-gui", daemon=True)
        self._thread.start()
        return self

    def stop(self) -> None:
        if self._httpd is not None:
            self._httpd.shutdown()
            self._httpd.server_close()
            self._httpd = None
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    # -- pipeline side -----------------------------------------------------
    def push_frame(self, stream_id: int, argb: np.ndarray) -> None:
        """Publish the latest ARGB32 [H][W] pixmap of one data stream
        (frames are BMP-encoded here, off the GET path)."""
        bmp = encode_bmp(np.asarray(argb, dtype=np.uint32))
        with self._lock:
            self._frames[stream_id] = bmp

    def push_spectrum(self, stream_id: int, values) -> None:
        """Publish the latest per-channel intensity line of one stream
        (the reference's spectrum.qml view); downsampled to <= 1024 points."""
        import numpy as np
        v = np.asarray(values, dtype=np.float64).ravel()
        if v.size > 1024:
            n = (v.size // 1024) * 1024
            v = v[:n].reshape(1024, -1).mean(axis=1)
        body = json.dumps({"values": [round(float(x), 6) for x in v]})
        with self._lock:
            self._spectra[stream_id] = body.encode()

    def update_status(self, **kv) -> None:
        with self._lock:
            self._status.update(kv)

    # -- HTTP side ---------------------------------------------------------
    def status(self) -> dict:
        with self._lock:
            return {"streams": sorted(self._frames.keys()), **self._status}

    def get_frame_bmp(self, stream_id: int) -> bytes | None:
        with self._lock:
            return self._frames.get(stream_id)

    def get_spectrum_json(self, stream_id: int) -> bytes | None:
        with self._lock:
            return self._spectra.get(stream_id)

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
        return False
