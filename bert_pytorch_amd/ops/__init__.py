"""Hand-written HIP (gfx950) kernel wrappers.

Policy: on a GPU tensor the in-repo HIP extension ``bert_pytorch_amd._C``
MUST be present — a missing extension raises instead of silently falling
back to eager PyTorch (so GPU runs always exercise the native kernels).
On CPU tensors every op runs an eager fp32 composite (the same composite
the GPU numerics tests compare the HIP kernels against).

Set ``BPA_FORCE_EAGER=1`` to force the eager path on GPU (debugging only).
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR: str | None = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from bert_pytorch_amd import _C  # noqa: PLC0415

        _EXT = _C
    except ImportError as e:  # pragma: no cover - depends on build state
        _EXT_ERR = str(e)
    return _EXT


def extension():
    """Return the HIP extension module, raising if unavailable."""
    ext = _try_load()
    if ext is None:
        raise RuntimeError(
            "bert_pytorch_amd._C (the gfx950 HIP extension) is not built. "
            "Run `python setup.py build_ext --inplace` "
            f"(import error: {_EXT_ERR})"
        )
    return ext


def has_extension() -> bool:
    return _try_load() is not None


def use_native(tensor: torch.Tensor) -> bool:
    """True when the HIP path must be used for this tensor.

    GPU tensor + extension present -> native. GPU tensor + extension
    missing -> RuntimeError (fail loud; see module docstring). CPU -> eager.
    """
    if not tensor.is_cuda:
        return False
    if os.environ.get("BPA_FORCE_EAGER") == "1":
        return False
    extension()  # raises if missing
    return True


from .layernorm import FusedLayerNorm, fused_layer_norm  # noqa: E402,F401
from .bias_act import fused_bias_gelu  # noqa: E402,F401
from .fused_residual import fused_bias_dropout_residual_ln  # noqa: E402,F401
from .embedding import fused_embedding_ln_dropout  # noqa: E402,F401
from .attention import fused_attention  # noqa: E402,F401
from .cross_entropy import fused_cross_entropy  # noqa: E402,F401
from .linear import fused_linear  # noqa: E402,F401
from .matmul import linear_nobias  # noqa: E402,F401
from .ffn import ffn_supported, fused_ffn  # noqa: E402,F401
from .mlm import mlm_decoder_loss  # noqa: E402,F401
