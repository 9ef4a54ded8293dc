"""srtb_amd — MI355X-native real-time radio-telescope backend.

A from-scratch CDNA4 (gfx950) redesign of the capabilities of
fxzjshm/simple-radio-telescope-backend: hand-written HIP kernels for the
unpack → R2C FFT → RFI → coherent-dedispersion → waterfall-FFT →
spectral-kurtosis → single-pulse-detection chain, hipFFT/rocFFT for the large
transforms, HIP streams/events for the async pipeline, and RCCL over xGMI for
multi-GPU stream sharding.

Layout:
  config      — srtb_config.cfg-compatible runtime configuration
  ref         — NumPy oracle implementations of every device op
  ops         — HIP kernel wrappers (fail loudly without the extension on GPU)
  pipeline    — CPU oracle pipeline + GPU engine frontend
  parallel    — multi-GPU (torch.distributed / RCCL) stream sharding
  io          — telescope packet formats, UDP ingest helpers, file writers
  utils       — expression evaluator, logging, misc
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
from .config import Config  # noqa: F401
