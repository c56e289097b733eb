"""Compute ops: HIP-native on GPU, pure-torch reference on CPU.

``persia_amd.ops.native`` — the HIP extension (``persia_amd._C``) — is
REQUIRED whenever embeddings live on a GPU device; ops raise instead of
silently falling back to eager torch there.  The torch implementations in
``persia_amd.ops.reference`` are the numerics oracle (fp32) used by tests and
by the CPU execution path.
"""

import torch


_NATIVE = None
_NATIVE_ERR = None


def native():
    """Import and return the HIP extension module, or raise loudly."""
    global _NATIVE, _NATIVE_ERR
    if _NATIVE is not None:
        return _NATIVE
    if _NATIVE_ERR is not None:
        raise _NATIVE_ERR
    try:
        from persia_amd import _C  # built in-tree by setup.py build_ext --inplace

        _NATIVE = _C
        return _NATIVE
    except ImportError as e:  # pragma: no cover
        _NATIVE_ERR = ImportError(
            "persia_amd._C (HIP extension) is not built. On a GPU machine this "
            "is required — build it with `python setup.py build_ext --inplace` "
            f"(original error: {e})"
        )
        raise _NATIVE_ERR


def native_available() -> bool:
    try:
        native()
        return True
    except ImportError:
        return False


def require_native_on_gpu(device: torch.device):
    if device.type == "cuda":
        native()  # raises with a clear message if missing
