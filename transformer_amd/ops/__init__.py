"""HIP kernel extension loader.

The compiled CDNA4 extension (`_tfmx_C*.so`, built in-tree by
`__graft_entry__.build()` / `python setup.py build_ext --inplace`) provides
every device op.  On a GPU box the extension is
MANDATORY: ops raise rather than silently falling back to eager PyTorch.
The CPU path (test oracle + toy-corpus training) uses ops/reference.py.
"""

from __future__ import annotations

import importlib
import os

_EXT = None
_EXT_ERR: str | None = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        _EXT = importlib.import_module("transformer_amd.ops._tfmx_C")
    except ImportError as e:  # keep the reason for the loud failure later
        _EXT_ERR = str(e)


def ext():
    """Return the compiled extension module, or raise loudly.

    Called on the GPU path only — never as a silent fallback gate."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "transformer_amd HIP extension (_tfmx_C) is not built/loadable; "
            "refusing to run a silent eager fallback on GPU. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_EXT_ERR}"
        )
    return _EXT


def has_ext() -> bool:
    _try_load()
    return _EXT is not None


from . import reference  # noqa: E402
from .functional import (  # noqa: E402
    linear,
    fused_attention,
    self_attention,
    cross_attention,
    residual_layernorm,
    dropout_residual_layernorm,
    embedding_scale_pe,
    embedding_scale_pe_at,
    dropout,
    masked_cross_entropy,
    masked_accuracy,
    masked_accuracy_counts,
    argmax_lastdim,
)
