"""Op dispatch: hand-written HIP/CDNA4 kernels on GPU, PyTorch fp32 on CPU.

The in-tree extension `_sonata_hip` (built by setup.py / __graft_entry__.build
with PYTORCH_ROCM_ARCH=gfx950) provides the GPU implementations.  On a GPU
box the HIP path is mandatory: if a tensor is on `cuda` and the extension is
missing, ops raise instead of silently falling back (the CPU/PyTorch path is
the numerics oracle, not a serving path).

Set SONATA_FORCE_TORCH=1 to force the PyTorch path everywhere (used by
parity tests to produce the reference output on GPU).
"""

from __future__ import annotations

import os

_EXT = None
_EXT_ERR: str = ""


def _try_load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None:
        return _EXT
    try:
        import torch  # noqa: F401
        import importlib

        _EXT = importlib.import_module("sonata_amd.ops._sonata_hip")
    except Exception as e:  # pragma: no cover - depends on build state
        _EXT = None
        _EXT_ERR = str(e)
    return _EXT


def hip_ext(required: bool = False):
    """Return the loaded HIP extension module (or None).

    required=True raises if unavailable — used on the GPU path so a missing
    native build fails loudly instead of silently running eager PyTorch."""
    ext = _try_load_extension()
    if required and ext is None:
        raise RuntimeError(
            "sonata_amd HIP extension (_sonata_hip) is not available on a GPU "
            f"path — build it with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_EXT_ERR}"
        )
    return ext


def use_hip(tensor) -> bool:
    """True when `tensor` lives on GPU and the HIP kernels must run."""
    if os.environ.get("SONATA_FORCE_TORCH", "0") == "1":
        return False
    return bool(tensor.is_cuda)


from .functional import (  # noqa: F401,E402
    layer_norm_ct,
    fused_gate,
    prior_sample,
    expand_states,
    leaky_conv1d,
    leaky_convtranspose1d,
    conv_mod,
    mask_tail_,
    leaky_conv1d_cl,
    leaky_convtranspose1d_cl,
    resblock_pair_cl,
    fused_gate_cl,
    depthwise_conv1d_cl,
    attn_relpos_cl,
    row_ln_cl,
)
