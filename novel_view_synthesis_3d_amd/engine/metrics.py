"""Training observability (the reference has a bare print — SURVEY.md §5.5)."""

from __future__ import annotations

import json
import os
import time
from collections import deque
from typing import Optional


class RateMeter:
    """Sliding-window items/sec."""

    def __init__(self, window: int = 50):
        self.times = deque(maxlen=window)
        self.counts = deque(maxlen=window)

    def update(self, n: int) -> None:
        self.times.append(time.perf_counter())
        self.counts.append(n)

    def rate(self) -> float:
        if len(self.times) < 2:
            return 0.0
        dt = self.times[-1] - self.times[0]
        return sum(list(self.counts)[1:]) / dt if dt > 0 else 0.0


class JsonlLogger:
    def __init__(self, path: Optional[str], enabled: bool = True):
        self.enabled = enabled and path is not None
        self.path = path
        if self.enabled:
            os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
            self._f = open(path, "a")

    def log(self, **kv) -> None:
        if self.enabled:
            kv.setdefault("time", time.time())
            self._f.write(json.dumps(kv) + "\n")
            self._f.flush()

    def close(self) -> None:
        if self.enabled:
            self._f.close()
