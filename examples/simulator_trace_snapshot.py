"""Event-driven simulate() with trace + memory-snapshot artifacts.

Parity target: /root/reference/examples/simulator_trace_snapshot.py.
"""
import argparse
import json
import os
import sys
from pathlib import Path

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--output", type=Path, default=Path("simulator_llama2_tiny_mi355x"))
    ap.add_argument("--no-merge-lanes", action="store_true")
    args = ap.parse_args()

    perf = PerfLLM()
    perf.configure(
        strategy_config=StrategyConfig.init_from_config_file(
            get_simu_strategy_config("tp1_pp2_dp4_mbs1")),
        model_config=ModelConfig.init_from_config_file(
            get_simu_model_config("llama2-tiny")),
        system_config=SystemConfig.init_from_config_file(
            get_simu_system_config("mi355x")),
    )
    perf.model_config.layer_num = 2
    perf.run_estimate()
    res = perf.simulate(str(args.output), merge_lanes=not args.no_merge_lanes)

    trace = json.loads((args.output / "tracing_logs.json").read_text())
    slices = [e for e in trace["traceEvents"] if e.get("ph") == "X"]
    print(json.dumps({
        "total_time_ms": res["total_time"],
        "event_count": len(trace["traceEvents"]),
        "slice_count": len(slices),
        "comm_slice_count": sum(1 for e in slices if e["cat"] == "comm"),
        "rank_count": len({e["pid"] for e in slices}),
        "peak_mem_gib": {str(r): round(v / 2**30, 3)
                         for r, v in res["peak_mem"].items()},
    }, indent=2))


if __name__ == "__main__":
    main()
