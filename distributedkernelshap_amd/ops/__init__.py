"""HIP extension loader.

The extension is built in-tree (``python setup.py build_ext --inplace``) so
the ``.so`` ships with the repo snapshot to GPU boxes. On a machine with a
visible GPU a missing extension is a hard error — the HIP path must never be
silently replaced by an eager fallback.
"""
from __future__ import annotations

_ext = None
_err: Exception | None = None


def load_extension():
    global _ext, _err
    if _ext is not None:
        return _ext
    try:
        import torch  # noqa: F401  (loads libc10/libtorch before the ext)

        from . import _kshap_hip  # type: ignore

        _ext = _kshap_hip
        return _ext
    except Exception as e:  # pragma: no cover - import failure path
        _err = e
        import torch

        if torch.cuda.is_available():
            raise ImportError(
                "distributedkernelshap_amd HIP extension (_kshap_hip) is not "
                "built but a GPU is present. Build it with "
                "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH="
                f"gfx950). Underlying error: {e!r}"
            ) from e
        raise


def extension_available() -> bool:
    try:
        load_extension()
        return True
    except Exception:
        return False
