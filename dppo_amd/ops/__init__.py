"""dppo_amd.ops — compute ops with eager-PyTorch reference paths and
hand-written CDNA4 HIP fast paths.

Every op has two implementations:
  * a pure-PyTorch reference (runs anywhere, used on CPU and as the
    numerics oracle in tests), and
  * a HIP/gfx950 kernel in dppo_amd/ops/hip/, compiled in-tree by
    `python -m dppo_amd.ops.build` (or __graft_entry__.build()).

Dispatch policy (config USE_HIP_KERNELS):
  'auto'   — HIP on CUDA tensors when the extension is importable;
             on a GPU machine a missing extension is an ERROR (we never
             silently fall back to eager on the GPU: that would make GPU
             tests pass on a path the benchmark does not use).
  'always' — HIP or raise.
  'never'  — eager everywhere (A/B testing).
"""

from __future__ import annotations

import os

import torch

from .gae import gae_advantages
from .ppo_loss import ppo_losses, PPOLossCoeffs

_ext = None
_ext_error: str | None = None


def hip_ext():
    """The compiled HIP extension module, or None on CPU-only hosts."""
    global _ext, _ext_error
    if _ext is not None:
        return _ext
    try:
        from . import _hip_loader

        _ext = _hip_loader.load()
    except Exception as e:  # noqa: BLE001
        _ext_error = f"{type(e).__name__}: {e}"
        _ext = None
    return _ext


def require_hip_ext():
    ext = hip_ext()
    if ext is None:
        raise RuntimeError(
            "dppo_amd HIP extension is not available on this GPU host "
            f"(load error: {_ext_error}). Build it in-tree with "
            "`python -m dppo_amd.ops.build` — eager fallback is disabled "
            "on GPU by design."
        )
    return ext


def use_hip(tensor_or_device, policy: str = "auto") -> bool:
    """Decide whether the HIP path should run for this tensor/device."""
    dev = (
        tensor_or_device.device
        if isinstance(tensor_or_device, torch.Tensor)
        else torch.device(tensor_or_device)
    )
    if policy == "never" or dev.type != "cuda":
        return False
    # On a GPU box the HIP extension must exist; fail loudly otherwise.
    require_hip_ext()
    return True


__all__ = [
    "gae_advantages",
    "ppo_losses",
    "PPOLossCoeffs",
    "hip_ext",
    "require_hip_ext",
    "use_hip",
]
