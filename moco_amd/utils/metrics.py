"""JSONL metrics sink (SURVEY §5.5 rebuild plan: stdout meters stay, plus a
machine-readable metrics stream for throughput/scaling curves)."""

import json
import os
import time
from typing import Optional


class MetricsLogger:
    """Append-only JSONL writer; no-op when path is None (non-master ranks)."""

    def __init__(self, path: Optional[str]):
        self.path = path
        self._f = None
        if path:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            self._f = open(path, "a")

    def log(self, **fields):
        if self._f is None:
            return
        fields.setdefault("ts", round(time.time(), 3))
        self._f.write(json.dumps(fields) + "\n")
        self._f.flush()

    def close(self):
        if self._f is not None:
            self._f.close()
            self._f = None
