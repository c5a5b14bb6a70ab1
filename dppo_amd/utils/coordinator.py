"""Lifecycle coordinator (tf.train.Coordinator parity).

The reference uses tf.train.Coordinator for stop/join (reference
main.py:32,62; Chief.py:20,87; Worker.py:30).  In the rebuild the stop
signal also crosses ranks (an all-gathered stop flag in the round
protocol), but the in-process surface is the same: should_stop(),
request_stop(), join().
"""

from __future__ import annotations

import threading
from typing import Iterable, Optional


class Coordinator:
    def __init__(self) -> None:
        self._stop = threading.Event()
        self._exc: Optional[BaseException] = None

    def should_stop(self) -> bool:
        return self._stop.is_set()

    def request_stop(self, exc: Optional[BaseException] = None) -> None:
        if exc is not None and self._exc is None:
            self._exc = exc
        self._stop.set()

    def join(self, threads: Iterable[threading.Thread] = (), timeout: Optional[float] = None) -> None:
        for t in threads:
            t.join(timeout)
        if self._exc is not None:
            raise self._exc
