"""Worker — rollout actor (API parity with reference Worker.py:8-153).

In the reference a Worker is a thread that owns one gym env and a full
policy-graph replica inside a shared tf.Session.  Here a Worker is the
per-process (= per-GPU) training role: it owns a batch of E device-resident
synthetic envs and a pi/oldpi replica, and participates in the synchronous
round protocol over RCCL.  `work()` is the reference's Worker.work loop
(Worker.py:29-138); `act(s)` is the single-state epsilon-greedy action
(Worker.py:140-153).
"""

from __future__ import annotations

from typing import Any, Dict, Optional, Tuple

import torch

from .config import DPPOConfig
from .parallel.comm import Comm
from .trainer import DPPOEngine
from .utils.coordinator import Coordinator


class Worker:
    def __init__(
        self,
        scope: str,
        parameter_dict: Any,
        session: Any = None,       # accepted for reference-signature parity
        memory_dict: Any = None,   # (Worker.py:9); unused — rollouts are
        coord: Optional[Coordinator] = None,  # rank-local, SURVEY.md §2.3
        comm: Optional[Comm] = None,
        engine: Optional[DPPOEngine] = None,
    ):
        if isinstance(parameter_dict, DPPOConfig):
            cfg = parameter_dict
        else:
            cfg = DPPOConfig.from_dict(dict(parameter_dict))
        self.cfg = cfg
        self.name = scope
        self.COORD = coord if coord is not None else Coordinator()
        self.engine = engine if engine is not None else DPPOEngine(cfg, comm=comm, scope=scope)
        self.CUR_EP = 0

    # ------------------------------------------------------------------
    def work(
        self,
        push_event: Any = None,    # reference signature parity (Worker.py:29);
        update_event: Any = None,  # the Event pair is replaced by the
        log_writer: Any = None,    # synchronous round protocol
        max_rounds: Optional[int] = None,
    ) -> Dict[str, float]:
        """Run training rounds until the global stop rule fires
        (Worker.py:30 while-loop + Chief stop, Chief.py:85-87)."""
        stats: Dict[str, float] = {}
        n = 0
        while not self.COORD.should_stop():
            stats, stop = self.engine.train_round()
            self.CUR_EP = self.engine.CUR_EP
            n += 1
            if stop or (max_rounds is not None and n >= max_rounds):
                self.COORD.request_stop()
        return stats

    # ------------------------------------------------------------------
    def act(self, s) -> Tuple[Any, float]:
        """Epsilon-greedy action + value for one state (Worker.py:140-153).

        Returns (action, pred_v) like the reference (action as python
        scalar for Discrete, numpy array for Box)."""
        eng = self.engine
        s_t = torch.as_tensor(s, device=eng.device, dtype=eng.dtype).unsqueeze(0)
        a, v, _ = eng.act_batch(s_t, eng.exploration_rate())
        a0 = a[0]
        if eng._discrete:
            return int(a0), float(v[0])
        return a0.detach().cpu().numpy(), float(v[0])
