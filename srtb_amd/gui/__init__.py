"""Live waterfall GUI (browser-based; reference gui/ Qt5 equivalent)."""

from .server import WaterfallServer, encode_bmp  # noqa: F401
