"""torch_util — parity surface for the reference's vendored tf_util
(reference Others/tf_util.py).

Only the LIVE and capability-relevant pieces are rebuilt (the reference
file is a 753-line grab-bag of which ~10 functions are reachable):

  reference (tf_util.py)        here
  ------------------------      ----------------------------------
  clip (:14)                    clip
  mean/sum/max/argmax (:17-51)  mean / sum / max / argmax
  normc_initializer (:286)      normc_initializer (models.mlp.normc_init_)
  save_state/load_state (:271)  re-exported from dppo_amd.checkpoint
  GetFlat (:633-643)            get_flat
  SetFromFlat (:609-631)        set_from_flat
  flatgrad (:600-607)           flatgrad
"""

from __future__ import annotations

from typing import Iterable, List, Optional, Sequence

import torch

from ..checkpoint import load_state, save_state  # noqa: F401  (re-export)
from ..models.mlp import normc_init_


def clip(x: torch.Tensor, lo, hi) -> torch.Tensor:
    """tf.clip_by_value (tf_util.py:14)."""
    return torch.clamp(x, lo, hi)


def mean(x: torch.Tensor, axis: Optional[int] = None, keepdims: bool = False):
    return x.mean() if axis is None else x.mean(dim=axis, keepdim=keepdims)


def sum(x: torch.Tensor, axis: Optional[int] = None, keepdims: bool = False):  # noqa: A001
    return x.sum() if axis is None else x.sum(dim=axis, keepdim=keepdims)


def max(x: torch.Tensor, axis: Optional[int] = None, keepdims: bool = False):  # noqa: A001
    return x.max() if axis is None else x.max(dim=axis, keepdim=keepdims).values


def argmax(x: torch.Tensor, axis: int):
    return torch.argmax(x, dim=axis)


def normc_initializer(std: float = 1.0):
    """Returns an initializer callable (tf_util.py:286-291)."""

    def _init(weight: torch.Tensor) -> torch.Tensor:
        return normc_init_(weight, std)

    return _init


# -- flat-vector parameter I/O (tf_util.py:600-643) ------------------------


def get_flat(params: Iterable[torch.Tensor]) -> torch.Tensor:
    """Concatenate parameters into one flat fp32 vector (GetFlat)."""
    return torch.cat([p.detach().reshape(-1) for p in params])


@torch.no_grad()
def set_from_flat(params: Iterable[torch.Tensor], flat: torch.Tensor) -> None:
    """Scatter a flat vector back into parameters (SetFromFlat)."""
    offset = 0
    for p in params:
        n = p.numel()
        p.copy_(flat[offset:offset + n].view_as(p))
        offset += n
    if offset != flat.numel():
        raise ValueError(f"flat vector has {flat.numel()} elements, params need {offset}")


def flatgrad(
    loss: torch.Tensor,
    params: Sequence[torch.Tensor],
    clip_norm: Optional[float] = None,
) -> torch.Tensor:
    """Flat gradient of loss wrt params, optional per-tensor norm clip
    (tf_util.py:600-607)."""
    grads = torch.autograd.grad(loss, list(params), allow_unused=True)
    out: List[torch.Tensor] = []
    for p, g in zip(params, grads):
        g = torch.zeros_like(p) if g is None else g
        if clip_norm is not None:
            norm = g.norm()
            if norm > clip_norm:
                g = g * (clip_norm / norm)
        out.append(g.reshape(-1))
    return torch.cat(out)
