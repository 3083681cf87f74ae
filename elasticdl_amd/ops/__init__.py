"""HIP kernel extension loader.

On a GPU box the native extension is mandatory — ops fail loudly rather
than silently falling back to eager torch. On CPU-only machines (unit
tests, the build container) the torch reference implementations in
``elasticdl_amd.ops.reference`` are used instead.
"""

import torch

try:
    from elasticdl_amd.ops import _C  # noqa: F401

    HAS_NATIVE = True
except ImportError as e:  # pragma: no cover - build environments
    _C = None
    HAS_NATIVE = False
    _IMPORT_ERROR = e


def native_available() -> bool:
    return HAS_NATIVE


def require_native():
    """Return the extension, refusing to run CUDA-device work without it."""
    if not HAS_NATIVE:
        raise RuntimeError(
            "elasticdl_amd.ops._C is not built — run "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`. "
            f"(import error: {_IMPORT_ERROR})"
        )
    return _C


def use_native(device) -> bool:
    """Native kernels run for CUDA tensors; CPU uses reference ops."""
    dev = torch.device(device)
    if dev.type == "cuda":
        require_native()
        return True
    return False
