"""Checkpoint / resume (capability from reference Others/tf_util.py:271-279).

The reference ships save_state/load_state (tf.train.Saver wrappers) but
never wires them.  The rebuild keeps the capability AND
the layout contract: a save addresses the full variable set by
scope-qualified names — pi, oldpi, and the Adam moments — so a run can
resume exactly.  Rank 0 writes; all ranks load (then params are already
identical; a defensive broadcast follows in load_state when a comm is
given).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from .trainer import DPPOEngine

FORMAT_VERSION = 1


def _scoped_state(module: torch.nn.Module, scope: str):
    # clone: the params are views into the flat buffer; saving a view
    # would serialize the whole flat storage once per variable.
    return {
        f"{scope}/{k}": v.detach().clone() for k, v in module.state_dict().items()
    }


def save_state(path: str, engine: DPPOEngine) -> Optional[str]:
    """Write the full training state; rank 0 only (returns path or None).

    makedirs like the reference's save_state (tf_util.py:274-276)."""
    if engine.comm.rank != 0:
        engine.comm.barrier()
        return None
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    payload = {
        "format_version": FORMAT_VERSION,
        "scope": engine.scope,
        # scope-qualified variable names: '<scope>pi/...' and
        # '<scope>oldpi/...', mirroring the reference's variable_scope
        # naming (PPO.py:21-22: scope+'pi', scope+'oldpi').
        "variables": {
            **_scoped_state(engine.pi, engine.scope + "pi"),
            **_scoped_state(engine.oldpi, engine.scope + "oldpi"),
        },
        "adam": engine.optimizer.state_dict(),
        "CUR_EP": engine.CUR_EP,
        "round": engine._round,
        "config": engine.cfg.to_dict(),
    }
    torch.save(payload, path)
    engine.comm.barrier()
    return path


def load_state(path: str, engine: DPPOEngine) -> None:
    """Restore pi, oldpi, Adam moments and progress counters on every rank."""
    payload = torch.load(path, map_location=engine.device, weights_only=False)
    if payload.get("format_version") != FORMAT_VERSION:
        raise ValueError(f"unknown checkpoint format: {payload.get('format_version')}")
    scope = payload["scope"]
    variables = payload["variables"]

    def unscope(prefix):
        plen = len(prefix) + 1
        return {k[plen:]: v for k, v in variables.items() if k.startswith(prefix + "/")}

    engine.pi.load_state_dict(unscope(scope + "pi"))
    engine.oldpi.load_state_dict(unscope(scope + "oldpi"))
    # load_state_dict copies INTO the flat-buffer views, so flat_param
    # stays the storage of record; refresh optimizer state after.
    engine.optimizer.load_state_dict(payload["adam"])
    engine.CUR_EP = int(payload["CUR_EP"])
    engine._round = int(payload["round"])
    # defensive drift guard: everyone loads the same file, but make the
    # invariant explicit
    engine.comm.broadcast_(engine.flat_pi.flat_param, src=0)
    engine.comm.broadcast_(engine.flat_old.flat_param, src=0)
    # the wide-config path caches bf16 weight copies keyed on a dirty
    # flag that updates set — a restore changes params outside that
    # flow, so invalidate explicitly (stale-weight rollout otherwise)
    wide = getattr(engine, "_wide_path", None)
    if wide is not None:
        wide.mark_dirty()
