"""HIP kernel dispatch layer.

Every hot op has two paths:
  * GPU (``tensor.is_cuda``): the hand-written gfx950 HIP kernel from the
    in-tree extension ``nanorlhf_amd/_C.so``.  If the extension is missing on
    a machine with a GPU this raises loudly — there is NO silent eager
    fallback on GPU.
  * CPU: a plain-PyTorch fp32 reference used by the CPU test suite and as
    the numerics oracle for the kernels.
"""
from __future__ import annotations

import importlib

import torch  # noqa: F401  (must load before _C: provides libc10)

_EXT = None
_EXT_ERR: Exception | None = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        _EXT = importlib.import_module("nanorlhf_amd._C")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e
        _EXT = None
    return _EXT


def ext():
    """The HIP extension module; raises loudly if unavailable."""
    m = _load()
    if m is None:
        raise RuntimeError(
            "nanorlhf_amd._C HIP extension is not built/importable but a GPU op "
            "was requested. Build it in-tree with `python setup.py build_ext "
            f"--inplace` (PYTORCH_ROCM_ARCH=gfx950). Original error: {_EXT_ERR!r}"
        )
    return m


def ext_available() -> bool:
    return _load() is not None


from .rmsnorm import add_rms_norm, rms_norm  # noqa: E402,F401
from .rope import build_rope_cache, rope_apply  # noqa: E402,F401
from .swiglu import swiglu  # noqa: E402,F401
from .attention import flash_attn_varlen  # noqa: E402,F401
from .logprob import token_logprob_entropy  # noqa: E402,F401
from .adamw import FusedAdamW  # noqa: E402,F401
from .sampling import sample_tokens  # noqa: E402,F401
from .kvcache import kv_append, paged_attn_decode  # noqa: E402,F401
