"""Chrome-trace export (Perfetto-compatible).

Parity target: simumax/core/generate_tracing.py (process_log_file ->
tracing_logs.json with ordered lanes, wait segments, flow arrows) and
trace_export.py (analytic schedules in the same format). Consumer
contract (examples/README.md in the reference): events carry
ph/ts/dur/pid="rankN"/tid/cat/args.call_stack/args.stream_type;
wrapper {"traceEvents": [...], "displayTimeUnit": "ms"}.
"""

from __future__ import annotations

import json

_LANE_ORDER = ["fwd compute", "bwd compute", "recompute", "optimizer",
               "comm", "wait", "pp"]


def _tid(kind: str, lane: str) -> str:
    if lane == "comm":
        return "pp" if kind == "p2p" else "comm"
    return {
        "fwd": "fwd compute",
        "bwd": "bwd compute",
        "recompute": "recompute",
        "optim": "optimizer",
    }.get(kind, "fwd compute")


def events_to_chrome_trace(log, save_path: str):
    """log: list of sim.events.LogEvent -> tracing_logs.json."""
    out = []
    # lane-order metadata so Perfetto displays lanes consistently
    ranks = sorted({e.rank for e in log})
    for r in ranks:
        out.append(dict(ph="M", name="process_name", pid=f"rank{r}",
                        args={"name": f"rank{r}"}))
        for i, lane in enumerate(_LANE_ORDER):
            out.append(dict(ph="M", name="thread_sort_index", pid=f"rank{r}",
                            tid=lane, args={"sort_index": i}))
    for e in log:
        tid = _tid(e.kind, e.lane)
        cat = "comm" if e.lane == "comm" else "compute"
        # explicit wait segment when the op sat blocked at lane head
        if e.wait_start < e.start - 1e-9:
            out.append(dict(
                ph="X", name=f"wait:{e.name}", pid=f"rank{e.rank}", tid="wait",
                ts=e.wait_start * 1e3, dur=(e.start - e.wait_start) * 1e3,
                cat="wait", args={"stream_type": "wait"}))
        out.append(dict(
            ph="X", name=e.name, pid=f"rank{e.rank}", tid=tid,
            ts=e.start * 1e3, dur=max(e.end - e.start, 0.0) * 1e3, cat=cat,
            args={"call_stack": e.call_stack or e.name,
                  "stream_type": "comm" if e.lane == "comm" else "compute",
                  "microbatch": e.mb}))
    # flow arrows sender-post -> receiver-transfer for every p2p rendezvous
    # (reference parity: generate_tracing.py flow events). The sender's
    # zero-duration post is the arrow tail; the receiver's transfer slice
    # (positive duration, same gid, other rank) is the head.
    flows = {}
    for e in log:
        if e.kind != "p2p" or not getattr(e, "gid", None):
            continue
        slot = flows.setdefault(e.gid, {})
        if "send" in e.name:            # sender post (zero duration)
            slot.setdefault("send", e)
        elif e.end > e.start:           # receiver transfer slice
            slot.setdefault("recv", e)
    flow_id = 0
    for gid, slot in flows.items():
        snd, rcv = slot.get("send"), slot.get("recv")
        if snd is None or rcv is None or snd.rank == rcv.rank:
            continue
        flow_id += 1
        out.append(dict(ph="s", id=flow_id, name="p2p", cat="flow",
                        pid=f"rank{snd.rank}", tid=_tid("p2p", "comm"),
                        ts=snd.start * 1e3))
        out.append(dict(ph="f", id=flow_id, bp="e", name="p2p", cat="flow",
                        pid=f"rank{rcv.rank}", tid=_tid("p2p", "comm"),
                        ts=rcv.end * 1e3))
    payload = {"traceEvents": out, "displayTimeUnit": "ms"}
    with open(save_path, "w") as f:
        json.dump(payload, f)
    return payload


def export_pipeline_schedule_trace(schedule_records, fwd, bwd, save_path: str):
    """Analytic 1F1B schedule (perf path) -> same Chrome-trace format.
    Parity: trace_export.py:104-125."""
    out = []
    for rec in schedule_records:
        out.append(dict(
            ph="X", name=f"{rec.kind}{rec.mb}", pid=f"rank{rec.stage}",
            tid="fwd compute" if rec.kind == "F" else "bwd compute",
            ts=rec.start * 1e3, dur=(rec.end - rec.start) * 1e3,
            cat="compute",
            args={"stream_type": "compute", "microbatch": rec.mb}))
    payload = {"traceEvents": out, "displayTimeUnit": "ms"}
    with open(save_path, "w") as f:
        json.dump(payload, f)
    return payload
