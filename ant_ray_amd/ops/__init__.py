"""ant_ray_amd.ops — hand-written CDNA4 kernels behind autograd functions.

GPU path: ant_ray_amd._hip_ops (built in-tree by csrc/build.py, gfx950 only).
CPU path: ops.reference (pure torch fp32) so tests run without a GPU.
On a GPU box a missing extension FAILS LOUDLY — there is no silent eager
fallback for CUDA tensors.
"""
from __future__ import annotations

import torch

from ant_ray_amd.ops import reference
from ant_ray_amd.ops.functional import (  # noqa: F401
    adamw_step,
    attention,
    attention_decode,
    cast_affine,
    chunked_prefill_attention,
    decode_step_attn,
    fused_add_rmsnorm,
    nhwc_to_nchw,
    rmsnorm,
    rope_attention,
    rope_qkv,
    swiglu,
)
from ant_ray_amd.ops.loss import linear_cross_entropy  # noqa: F401
from ant_ray_amd.ops.reference import rope_tables  # noqa: F401

_hip = None
_hip_err = None
try:
    from ant_ray_amd import _hip_ops as _hip  # noqa: F401
except ImportError as e:  # pragma: no cover
    _hip_err = e


def hip_ops():
    """The HIP extension module; raises if absent (no silent fallback)."""
    if _hip is None:
        raise RuntimeError(
            "ant_ray_amd._hip_ops is not built. Run `python ant_ray_amd/csrc/build.py` "
            f"(import error: {_hip_err})"
        )
    return _hip


def have_hip() -> bool:
    return _hip is not None
