"""Chief — trainer / parameter-server role (API parity with reference
Chief.py:8-92).

In the reference the Chief owns the canonical weights and the only live
Adam, gates on all-workers-ready, averages tower gradients in-graph and
broadcasts weights back.  In the MI355X rebuild those responsibilities
are distributed: every rank runs the identical all-reduced update, so the
canonical weights live (bit-identically) on every rank, and the Chief is
rank 0's view of the run — it owns the logging, the periodic drift-guard
broadcast source, and the eval-time `act`.

`check()` is the reference's Chief.check loop (Chief.py:19-28).  Because
the rebuild's round protocol is synchronous, check() and Worker.work()
are the same loop observed from the two roles: call exactly one of them
per process.  When constructed standalone (tests, single process), the
Chief builds its own engine.
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional

import numpy as np

from .config import DPPOConfig
from .parallel.comm import Comm
from .trainer import DPPOEngine
from .utils.coordinator import Coordinator
from .worker import Worker


class Chief:
    def __init__(
        self,
        scope: str,
        parameter_dict: Any,
        session: Any = None,      # reference-signature parity (Chief.py:9)
        memory_dict: Any = None,
        coord: Optional[Coordinator] = None,
        workers: Optional[List[Worker]] = None,
        comm: Optional[Comm] = None,
    ):
        if isinstance(parameter_dict, DPPOConfig):
            cfg = parameter_dict
        else:
            cfg = DPPOConfig.from_dict(dict(parameter_dict))
        self.cfg = cfg
        self.name = scope
        self.workers = workers or []
        self.COORD = coord if coord is not None else (
            self.workers[0].COORD if self.workers else Coordinator()
        )
        # The Chief shares the local worker's engine (the reference shares
        # one tf graph across roles); standalone it builds its own.
        if self.workers:
            self.engine = self.workers[0].engine
        else:
            self.engine = DPPOEngine(cfg, comm=comm, scope=scope)
        self.UPDATE_STEPS = cfg.UPDATE_STEPS

    # ------------------------------------------------------------------
    def check(
        self,
        push_event: Any = None,
        update_event: Any = None,
        max_rounds: Optional[int] = None,
    ) -> Dict[str, float]:
        """Drive training rounds (Chief.py:19-28).  Synchronous analog of
        the Event-gated loop: each train_round gathers every rank's batch
        stats (the all-ready barrier, Chief.py:22-23), updates, and
        applies the stop rule."""
        stats: Dict[str, float] = {}
        n = 0
        while not self.COORD.should_stop():
            stats, stop = self.engine.train_round()
            n += 1
            if stop or (max_rounds is not None and n >= max_rounds):
                self.COORD.request_stop()
        return stats

    # ------------------------------------------------------------------
    def act(self, s):
        """Sample an action from the canonical policy, no exploration
        overlay (Chief.py:89-92 — note the reference samples rather than
        taking the mode)."""
        a = self.engine.act(np.asarray(s))
        if self.engine._discrete:
            return int(a)
        return a.detach().cpu().numpy()
