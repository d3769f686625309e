"""CLI front-end (the streamlit GUI analog of the reference's
app/streamlit_app.py — this image has no streamlit, so the primary UX is
`python -m simumax_amd <cmd>`; app/streamlit_app.py wraps the same calls
when streamlit is available).

Commands:
  analyze   perf estimate: MFU / iter time / peak memory (+ artifacts)
  simulate  event-driven replay with Chrome trace + memory snapshot
  search    grid-search parallel strategies, rank by MFU
  capture   operator-graph JSON + Graphviz dot
  configs   list shipped model/strategy/system configs
"""

from __future__ import annotations

import argparse
import json
import sys

from . import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
               get_simu_model_config, get_simu_strategy_config,
               get_simu_system_config, show_simu_model_configs,
               show_simu_strategy_configs, show_simu_system_configs)


def _load(args):
    p = PerfLLM()
    st = StrategyConfig.init_from_config_file(
        get_simu_strategy_config(args.strategy))
    for kv in args.set or []:
        k, v = kv.split("=", 1)
        cur = getattr(st, k)
        if isinstance(cur, bool):
            v = v.lower() in ("1", "true", "yes")
        elif isinstance(cur, int):
            v = int(v)
        elif isinstance(cur, float):
            v = float(v)
        setattr(st, k, v)
    p.configure(
        st,
        ModelConfig.init_from_config_file(get_simu_model_config(args.model)),
        SystemConfig.init_from_config_file(get_simu_system_config(args.system)),
    )
    return p


def cmd_analyze(args):
    p = _load(args)
    p.run_estimate()
    p.analysis(args.output)


def cmd_simulate(args):
    p = _load(args)
    p.run_estimate()
    res = p.simulate(args.output or "simu_out", merge_lanes=not args.all_ranks)
    print(json.dumps({"total_time_ms": res["total_time"],
                      "trace": res["trace_path"],
                      "peak_mem_gib": {str(k): round(v / 2**30, 2)
                                       for k, v in res["peak_mem"].items()}},
                     indent=2))


def cmd_search(args):
    from .tuning.strategy_searcher import SearchSpace, StrategySearcher

    st = StrategyConfig.init_from_config_file(
        get_simu_strategy_config(args.strategy))
    searcher = StrategySearcher(
        ModelConfig.init_from_config_file(get_simu_model_config(args.model)),
        SystemConfig.init_from_config_file(get_simu_system_config(args.system)),
        st)
    res = searcher.search(args.world_size, args.global_batch_size,
                          SearchSpace(), verbose=True)
    for row in res.top(args.top):
        print(f"MFU {row['mfu']*100:6.2f}%  {row['parallelism']} "
              f"rc={row['recompute']} mbs={row['mbs']}")


def cmd_capture(args):
    from .core.graph import capture_graph

    p = _load(args)
    builder = capture_graph(p, save_prefix=args.output or "simu_graph")
    print(f"captured {len(builder.nodes)} ops -> "
          f"{args.output or 'simu_graph'}.json/.dot")


def cmd_configs(args):
    print("models:  ", ", ".join(show_simu_model_configs()))
    print("strategy:", ", ".join(show_simu_strategy_configs()))
    print("system:  ", ", ".join(show_simu_system_configs()))


def cmd_serve(args):
    import uvicorn

    from .webapp import app as webapp

    uvicorn.run(webapp, host=args.host, port=args.port)


def main(argv=None):
    ap = argparse.ArgumentParser(prog="simumax_amd", description=__doc__,
                                 formatter_class=argparse.RawDescriptionHelpFormatter)
    sub = ap.add_subparsers(dest="cmd", required=True)

    def common(sp):
        sp.add_argument("--model", default="llama3-8b")
        sp.add_argument("--strategy", default="tp1_pp1_dp8_mbs1")
        sp.add_argument("--system", default="mi355x")
        sp.add_argument("--set", action="append", metavar="KEY=VALUE",
                        help="override a strategy field")
        sp.add_argument("--output", default=None)

    for name, fn in (("analyze", cmd_analyze), ("simulate", cmd_simulate),
                     ("capture", cmd_capture)):
        sp = sub.add_parser(name)
        common(sp)
        if name == "simulate":
            sp.add_argument("--all-ranks", action="store_true")
        sp.set_defaults(fn=fn)
    sp = sub.add_parser("search")
    common(sp)
    sp.add_argument("--world-size", type=int, default=8)
    sp.add_argument("--global-batch-size", type=int, default=32)
    sp.add_argument("--top", type=int, default=10)
    sp.set_defaults(fn=cmd_search)
    sp = sub.add_parser("configs")
    sp.set_defaults(fn=cmd_configs)
    sp = sub.add_parser("serve", help="launch the web front-end")
    sp.add_argument("--host", default="127.0.0.1")
    sp.add_argument("--port", type=int, default=8642)
    sp.set_defaults(fn=cmd_serve)

    args = ap.parse_args(argv)
    args.fn(args)


if __name__ == "__main__":
    main()
