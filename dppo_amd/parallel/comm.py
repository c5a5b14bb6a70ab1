"""Distributed communication fabric: RCCL over xGMI (gloo on CPU).

This replaces the reference's in-graph DP backend:
  - in-graph expand_dims/concat/reduce_mean gradient averaging across
    towers (reference PPO.py:55-65)      -> all_reduce(AVG) on ONE flat
    fp32 gradient bucket.  The models here are small (KB..100 MB), and
    xGMI collectives at this size are latency-bound, so a single fused
    bucket beats bandwidth-tuned multi-bucket schemes; the wide config
    (BASELINE.json #5, ~100 MB bf16 grads) still fits one bucket per
    launch over 7x153 GB/s links.
  - assign-op weight broadcast (reference main.py:48-50, Chief.py:67-70)
    -> rank-0 broadcast of the flat parameter buffer.  Because gradients
    are all-reduced and Adam state is identical, replicas stay
    bit-identical; the broadcast remains as an init step and a periodic
    drift guard.
  - CPython deque rollout hand-off + score sort (reference Worker.py:136,
    Chief.py:33-53) -> all_gather of the per-rank stats row (the 9-float
    logs vector + cur_lr + validity), which is all that must cross ranks:
    batches stay rank-local (SURVEY.md §2.3 data plane).

One process per GPU; backend "nccl" IS RCCL on ROCm. Single-process mode
(world_size == 1) short-circuits every collective.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional

import torch
import torch.distributed as dist


class FlatBuffers:
    """Flat parameter + gradient buffers backing a module's parameters.

    Re-points every nn.Parameter's storage into one contiguous flat
    tensor and pre-assigns every .grad as a view into one flat gradient
    tensor, so that:
      - gradient all-reduce is ONE collective on `flat_grad`,
      - the optimizer runs ONE fused update on `flat_param`/`flat_grad`,
      - parameter broadcast / oldpi<-pi sync is ONE copy_.
    """

    def __init__(self, module: torch.nn.Module):
        params = [p for p in module.parameters() if p.requires_grad]
        if not params:
            raise ValueError("module has no trainable parameters")
        numels = [p.numel() for p in params]
        total = sum(numels)
        device, dtype = params[0].device, params[0].dtype
        self.flat_param = torch.zeros(total, device=device, dtype=dtype)
        self.flat_grad = torch.zeros(total, device=device, dtype=dtype)
        offset = 0
        self.params = params
        self.slices: List[slice] = []
        for p in params:
            n = p.numel()
            sl = slice(offset, offset + n)
            self.flat_param[sl].copy_(p.data.reshape(-1))
            p.data = self.flat_param[sl].view_as(p.data)
            p.grad = self.flat_grad[sl].view_as(p.data)
            self.slices.append(sl)
            offset += n
        self.numel = total
        # the flat master participates in the optimizer directly:
        # autograd accumulates into the per-param views of flat_grad, and
        # the optimizer consumes the whole bucket through flat_param.grad.
        self.flat_param.requires_grad_(True)
        self.flat_param.grad = self.flat_grad

    def zero_grad(self) -> None:
        self.flat_grad.zero_()

    @torch.no_grad()
    def copy_params_from(self, other: "FlatBuffers") -> None:
        self.flat_param.copy_(other.flat_param)


class Comm:
    """Process-group wrapper; world_size==1 short-circuits collectives."""

    def __init__(
        self,
        backend: Optional[str] = None,
        device: Optional[str] = None,
        timeout_s: float = 600.0,
    ):
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", str(self.rank)))
        self.distributed = self.world_size > 1

        if device is None:
            if torch.cuda.is_available():
                torch.cuda.set_device(self.local_rank % torch.cuda.device_count())
                device = f"cuda:{torch.cuda.current_device()}"
            else:
                device = "cpu"
        self.device = torch.device(device)

        if self.distributed and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if self.device.type == "cuda" else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            dist.init_process_group(
                backend=backend,
                rank=self.rank,
                world_size=self.world_size,
                timeout=datetime.timedelta(seconds=timeout_s),
            )
        self.backend = dist.get_backend() if self.distributed else "local"

    # -- collectives ---------------------------------------------------
    @torch.no_grad()
    def allreduce_mean_(self, flat: torch.Tensor) -> torch.Tensor:
        """In-place mean over ranks of one flat bucket (PPO.py:55-65 analog)."""
        if self.distributed:
            dist.all_reduce(flat, op=dist.ReduceOp.SUM)
            flat.div_(self.world_size)
        return flat

    @torch.no_grad()
    def broadcast_(self, flat: torch.Tensor, src: int = 0) -> torch.Tensor:
        """Rank-src parameter broadcast (main.py:48-50 / Chief.py:67-70 analog)."""
        if self.distributed:
            dist.broadcast(flat, src=src)
        return flat

    def all_gather_rows(self, row: torch.Tensor) -> torch.Tensor:
        """Gather a [k] stats row from every rank -> [world_size, k].

        Carries the reference's Chief control data across ranks: the
        9-float logs vector, cur_lr and the validity flag (SURVEY.md §2.3
        collective-call-site table, last row).
        """
        if not self.distributed:
            return row.unsqueeze(0)
        row = row.contiguous()
        # list-form all_gather: supported by both gloo and nccl/RCCL; the
        # row is a handful of floats so this is latency-, not copy-bound.
        outs = [torch.empty_like(row) for _ in range(self.world_size)]
        dist.all_gather(outs, row)
        return torch.stack(outs, dim=0)

    def send(self, t: torch.Tensor, dst: int) -> None:
        """Blocking point-to-point send (batch-curation transport)."""
        dist.send(t.contiguous(), dst=dst)

    def recv(self, t: torch.Tensor, src: int) -> torch.Tensor:
        """Blocking point-to-point receive into `t` (must be contiguous)."""
        dist.recv(t, src=src)
        return t

    def barrier(self) -> None:
        if self.distributed:
            dist.barrier()

    def shutdown(self) -> None:
        if self.distributed and dist.is_initialized():
            dist.destroy_process_group()
