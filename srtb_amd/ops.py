"""Access to the native HIP extension (srtb_amd._C).

On a GPU box the HIP path is mandatory: if the extension is missing or fails
to import while CUDA/HIP devices are visible, we raise — there is NO silent
eager/PyTorch fallback for device tensors.  CPU tensors are served by the
NumPy oracle (:mod:`srtb_amd.ref`) explicitly, for tests only.
"""

from __future__ import annotations

_ext = None
_ext_err: Exception | None = None


def native():
    """Return the native extension module, importing it on first use."""
    global _ext, _ext_err
    if _ext is not None:
        return _ext
    if _ext_err is not None:
        raise RuntimeError(
            "srtb_amd._C failed to import earlier; GPU ops unavailable"
        ) from _ext_err
    try:
        import torch  # noqa: F401 — loads libc10 etc. before the extension
        from srtb_amd import _C
        _ext = _C
        return _ext
    except Exception as e:  # noqa: BLE001
        _ext_err = e
        import torch
        if torch.cuda.is_available():
            raise RuntimeError(
                "GPU present but the srtb_amd HIP extension failed to "
                f"import ({e!r}). Run `python setup.py build_ext --inplace` "
                "(gfx950). Refusing to fall back to a non-HIP path on a GPU "
                "box.") from e
        raise


def has_native() -> bool:
    try:
        native()
        return True
    except Exception:  # noqa: BLE001
        return False
