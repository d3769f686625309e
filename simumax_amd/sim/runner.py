"""run_simulation: event-driven replay -> trace + memory artifacts.

Parity target: simumax/core/simu_runner.py:22-94 + simu_artifacts.py.
Artifacts written to save_path: tracing_logs.json, simu_memory_result.json,
simu_memory_snapshot.json, simu_memory_viz_snapshot.pickle.
"""

from __future__ import annotations

import os

from ..core.records import Result
from .events import SimuSystem
from .memory import SimuMemoryTracker
from .schedule import PpSchedule
from .trace import events_to_chrome_trace


def run_simulation(perf_model, save_path: str, merge_lanes: bool = True) -> Result:
    assert perf_model._estimated, "call run_estimate() first"
    os.makedirs(save_path, exist_ok=True)
    sched = PpSchedule(perf_model, merge_lanes=merge_lanes)
    jobs = sched.build()
    system = SimuSystem(jobs)
    total_ms = system.run()
    system.log.sort(key=lambda e: (e.rank, e.start))

    # memory replay (attached when pp==1 or sync pp — always true here,
    # reference parity simu_artifacts.py:9-10)
    s = perf_model.strategy
    per_stage = s.world_size // s.pp_size
    base = {}
    for r in jobs:
        stage = r // per_stage
        chunk = perf_model.chunks[stage]
        base[r] = chunk.get_model_info().all_bytes
    tracker = SimuMemoryTracker(base)
    for e in sorted(system.log, key=lambda x: (x.start, x.rank)):
        tracker.record(e.rank, e.start, e.name, e.kind, e.mem)

    trace_path = os.path.join(save_path, "tracing_logs.json")
    events_to_chrome_trace(system.log, trace_path)
    tracker.save(save_path)

    res = Result()
    res["total_time"] = total_ms
    res["trace_path"] = trace_path
    res["ranks"] = sorted(jobs)
    res["peak_mem"] = tracker.peak
    return res
