"""Metrics: rank-0 console logging + JSONL dump (SURVEY §5 observability).

The headline metric is images/sec over the whole job (BASELINE.json)."""

from __future__ import annotations

import json
import time


class Meter:
    def __init__(self, jsonl_path: str | None = None, rank: int = 0):
        self.rank = rank
        self.f = open(jsonl_path, "a") if (jsonl_path and rank == 0) else None
        self.t0 = time.perf_counter()

    def log(self, step: int, **scalars) -> None:
        if self.rank != 0:
            return
        rec = {"step": step, "t": time.perf_counter() - self.t0}
        rec.update({k: (float(v) if hasattr(v, "__float__") else v) for k, v in scalars.items()})
        line = " ".join(f"{k}={v:.5g}" if isinstance(v, float) else f"{k}={v}" for k, v in rec.items())
        print(line, flush=True)
        if self.f:
            self.f.write(json.dumps(rec) + "\n")
            self.f.flush()

    def close(self):
        if self.f:
            self.f.close()
