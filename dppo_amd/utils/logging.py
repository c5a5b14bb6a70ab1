"""Scalar logging (the reference's TB summaries, C9 in SURVEY.md §2.1).

The reference writes 4 loss scalars per round from Worker_N0 only
(reference PPO.py:41-45, Worker.py:112-114).  TensorBoard is not
installed in this environment, so the logger writes JSONL (one line per
step, trivially plottable) and mirrors to tensorboard.SummaryWriter when
that package exists.  Rank 0 logs; other ranks no-op, matching the
reference's single-writer discipline.
"""

from __future__ import annotations

import json
import os
import time
from typing import Dict


class ScalarLogger:
    def __init__(self, logdir: str, enabled: bool = True, name: str = "scalars"):
        self.enabled = enabled
        self.logdir = logdir
        self._fh = None
        self._tb = None
        if not enabled:
            return
        os.makedirs(logdir, exist_ok=True)
        self._fh = open(os.path.join(logdir, f"{name}.jsonl"), "a", buffering=1)
        try:  # optional TB mirror
            from torch.utils.tensorboard import SummaryWriter  # type: ignore

            self._tb = SummaryWriter(logdir)
        except Exception:  # noqa: BLE001
            self._tb = None

    def log(self, step: int, scalars: Dict[str, float]) -> None:
        if not self.enabled or self._fh is None:
            return
        rec = {"step": int(step), "time": time.time()}
        rec.update({k: float(v) for k, v in scalars.items()})
        self._fh.write(json.dumps(rec) + "\n")
        if self._tb is not None:
            for k, v in scalars.items():
                self._tb.add_scalar(k, float(v), step)

    def close(self) -> None:
        if self._fh is not None:
            self._fh.close()
            self._fh = None
        if self._tb is not None:
            self._tb.close()
            self._tb = None
