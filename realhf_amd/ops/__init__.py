"""Hot-op dispatch layer.

Every op has (a) a hand-written HIP/CDNA4 kernel in realhf_amd/ops/csrc
(built as the in-tree extension ``realhf_amd._C``) and (b) a pure-torch
reference used on CPU and as the numerics oracle in tests.

On a GPU box the HIP path is mandatory: if a CUDA tensor reaches an op and
the extension is missing, we raise — a silent eager fallback would
invalidate every benchmark (see repo rules).
Set REALHF_AMD_FORCE_TORCH_OPS=1 to explicitly allow the torch path on
GPU (numerics A/B testing only).
"""
import os

import torch

_C = None
_C_ERR = None
try:
    from realhf_amd import _C as _C  # built by setup.py build_ext --inplace
except ImportError as e:  # pragma: no cover
    _C_ERR = e


def hip_available() -> bool:
    return _C is not None


def require_hip():
    if _C is None:
        if os.environ.get("REALHF_AMD_FORCE_TORCH_OPS") == "1":
            return None
        raise RuntimeError(
            "realhf_amd._C HIP extension is not built but a CUDA tensor "
            f"reached a hot op (import error: {_C_ERR}). Build with "
            "`python setup.py build_ext --inplace`."
        )
    return _C


def use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if os.environ.get("REALHF_AMD_FORCE_TORCH_OPS") == "1" and _C is None:
        return False
    require_hip()
    return True
