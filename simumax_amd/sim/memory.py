"""SimuMemoryTracker: rank-local allocated-bytes timeline with cache
tokens (alloc/free pairing, size-checked), snapshots and a viz pickle.

Parity target: simumax/core/simu_memory.py:8-556 (OpMemoryProfile,
SimuMemoryTracker) and the artifact formats consumed by
examples/simulator_trace_snapshot.py (simu_memory_result.json,
simu_memory_snapshot.json, simu_memory_viz_snapshot.pickle).
"""

from __future__ import annotations

import json
import pickle
from collections import defaultdict
from typing import Dict, List


class MemoryTokenError(RuntimeError):
    pass


class SimuMemoryTracker:
    """Replays the simulator log's MemDelta records into per-rank
    allocated-bytes timelines with token-paired alloc/free."""

    SCHEMA = "simumax_amd.memory_snapshot.v1"

    def __init__(self, base_bytes_per_rank: Dict[int, float]):
        self.base = dict(base_bytes_per_rank)  # weights+grads+states
        self.allocated = dict(base_bytes_per_rank)
        self.live_tokens: Dict[int, Dict[str, float]] = defaultdict(dict)
        self.events: List[dict] = []
        self.cache_tokens: List[dict] = []
        self.peak: Dict[int, float] = dict(base_bytes_per_rank)
        self.peak_event: Dict[int, dict] = {}

    def record(self, rank, ts, op_name, phase, delta):
        """delta: events.MemDelta (may be None)."""
        if delta is None:
            self._log_point(rank, ts, op_name, phase)
            return
        if delta.alloc_bytes:
            key = delta.token_key or op_name
            if key in self.live_tokens[rank]:
                raise MemoryTokenError(f"double alloc of token {key} on rank {rank}")
            self.live_tokens[rank][key] = delta.alloc_bytes
            self.allocated[rank] += delta.alloc_bytes
            self.cache_tokens.append(dict(action="alloc", key=key, rank=rank,
                                          size=delta.alloc_bytes, ts=ts))
        peak_here = self.allocated[rank] + (delta.transient_bytes or 0.0)
        if peak_here > self.peak.get(rank, 0.0):
            self.peak[rank] = peak_here
            self.peak_event[rank] = dict(rank=rank, op_name=op_name,
                                         phase=phase, ts=ts,
                                         allocated_bytes=peak_here)
        self._log_point(rank, ts, op_name, phase,
                        transient=delta.transient_bytes)
        if delta.free_bytes:
            key = delta.token_key or op_name
            live = self.live_tokens[rank]
            if key in live:
                size = live.pop(key)
                if abs(size - delta.free_bytes) > max(1.0, 0.01 * size):
                    raise MemoryTokenError(
                        f"token {key} freed with size {delta.free_bytes} "
                        f"but allocated {size}")
                self.allocated[rank] -= size
            else:
                self.allocated[rank] -= delta.free_bytes
            self.cache_tokens.append(dict(action="free", key=key, rank=rank,
                                          size=delta.free_bytes, ts=ts))

    def _log_point(self, rank, ts, op_name, phase, transient=0.0):
        self.events.append(dict(rank=rank, ts=ts, op_name=op_name, phase=phase,
                                allocated_bytes=self.allocated[rank] + (transient or 0.0)))

    # ---- checks ----------------------------------------------------------
    def assert_balanced(self):
        for rank, live in self.live_tokens.items():
            if live:
                raise MemoryTokenError(
                    f"rank {rank}: {len(live)} cache tokens never freed, "
                    f"e.g. {list(live)[:3]}")

    # ---- artifacts -------------------------------------------------------
    def result(self) -> dict:
        return {
            f"rank{r}": {
                "base_model_mem": self.base[r],
                "peak_allocated_mem": self.peak[r],
                "peak_point": self.peak_event.get(r, {}).get("op_name", ""),
            }
            for r in sorted(self.base)
        }

    def snapshot(self) -> dict:
        return {
            "schema": self.SCHEMA,
            "events": self.events,
            "cache_tokens": self.cache_tokens,
        }

    def save(self, save_dir: str):
        import os

        with open(os.path.join(save_dir, "simu_memory_result.json"), "w") as f:
            json.dump(self.result(), f, indent=2)
        with open(os.path.join(save_dir, "simu_memory_snapshot.json"), "w") as f:
            json.dump(self.snapshot(), f)
        with open(os.path.join(save_dir, "simu_memory_viz_snapshot.pickle"), "wb") as f:
            pickle.dump({
                "schema": self.SCHEMA,
                "timelines": {
                    r: [(e["ts"], e["allocated_bytes"]) for e in self.events
                        if e["rank"] == r]
                    for r in sorted(self.base)
                },
            }, f)
