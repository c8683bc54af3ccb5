"""Loader for the native extension (CPU reference ops + gfx950 HIP kernels).

The extension is built in-tree (``python setup.py build_ext --inplace``) so
the .so travels with the repo snapshot to GPU boxes.  On a GPU box the HIP
path must be the one that runs: ops fail loudly if the extension is missing
rather than falling back to eager PyTorch.
"""
import torch

_C = None
_IMPORT_ERROR = None

try:
    from parallel_cnn_amd import _C as _ext  # type: ignore

    _C = _ext
except ImportError as e:  # pragma: no cover - exercised only on broken builds
    _IMPORT_ERROR = e


def available() -> bool:
    return _C is not None


def require():
    """Return the native module, raising a loud error if it is missing."""
    if _C is None:
        raise RuntimeError(
            "parallel_cnn_amd._C native extension is not built. "
            "Run `python setup.py build_ext --inplace` at the repo root. "
            f"(import error: {_IMPORT_ERROR})"
        )
    return _C


def current_stream_handle() -> int:
    """Opaque HIP stream handle of torch's current stream (0 if no GPU)."""
    if torch.cuda.is_available():
        return torch.cuda.current_stream().cuda_stream
    return 0
