"""Closed-form dual-pipeline (DualPipe-like) duration / MFU calculator.

Parity target: simumax/pp_simu/utils.py:4-162 (duration_dualpp,
perf_dualpp): a standalone what-if calculator for the dual-pipeline
schedule where two half-batches flow in opposite directions and fwd/bwd
of different directions overlap on each stage.
"""

from __future__ import annotations


def duration_1f1b(pp: int, mbc: int, f: float, b: float, p2p: float = 0.0):
    """Classic 1F1B: (mbc + pp - 1) * (f + b) + 2*(pp-1)*p2p."""
    return (mbc + pp - 1) * (f + b) + 2 * (pp - 1) * p2p


def duration_dualpp(pp: int, mbc: int, f: float, b: float, p2p: float = 0.0,
                    overlap_ratio: float = 1.0):
    """Dual pipeline: two directions of mbc/2 microbatches each; in steady
    state a stage overlaps one direction's fwd with the other's bwd.
    overlap_ratio in [0,1]: 1 = perfect fwd/bwd overlap of the two streams.
    Returns total duration per iteration."""
    assert mbc % 2 == 0, "dual pipeline needs an even microbatch count"
    half = mbc // 2
    ideal = half * (f + b)  # each direction's serialized work on a stage
    steady = ideal * (2 - overlap_ratio)  # overlapped fraction collapses
    bubble = (pp - 1) * (f + b) / 2 + 2 * (pp - 1) * p2p
    return steady + bubble


def perf_dualpp(pp: int, mbc: int, f: float, b: float, flops_per_mb: float,
                peak_flops: float, p2p: float = 0.0, overlap_ratio: float = 1.0):
    """MFU of 1F1B vs dual-pipeline at the same per-stage costs."""
    t1 = duration_1f1b(pp, mbc, f, b, p2p)
    t2 = duration_dualpp(pp, mbc, f, b, p2p, overlap_ratio)
    total = flops_per_mb * mbc
    return {
        "mfu_1f1b": total / (t1 / 1e3) / peak_flops,
        "mfu_dualpp": total / (t2 / 1e3) / peak_flops,
        "duration_1f1b": t1,
        "duration_dualpp": t2,
        "speedup": t1 / t2,
    }


def dualpp_schedule_events(pp: int, mbc: int, f: float, b: float,
                           p2p: float = 0.0, overlap_ratio: float = 1.0):
    """Synthesize the dual-pipeline per-stage timeline (two directions of
    mbc/2 microbatches; steady-state fwd/bwd of opposite directions
    overlap) as LogEvents for the Chrome-trace exporter — the MI355X
    analog of the reference's matplotlib Gantt plots
    (pp_simu/utils.py: perf_dualpp figures)."""
    from ..sim.events import LogEvent

    assert mbc % 2 == 0
    half = mbc // 2
    log = []
    for stage in range(pp):
        # direction A enters at stage, direction B at pp-1-stage
        lead_a = stage * (f + p2p)
        lead_b = (pp - 1 - stage) * (f + p2p)
        t_a, t_b = lead_a, lead_b
        for m in range(half):
            log.append(LogEvent(stage, f"fwdA.mb{m}", "fwd", "comp",
                                t_a, t_a + f, m))
            t_a += f
            log.append(LogEvent(stage, f"fwdB.mb{m}", "fwd", "comm",
                                t_b, t_b + f, m))
            t_b += f
        # steady bwd: opposite-direction bwd overlaps by overlap_ratio
        start = max(t_a, t_b)
        step = (f + b) * (2 - overlap_ratio) / 2
        for m in range(half):
            log.append(LogEvent(stage, f"bwdA.mb{m}", "bwd", "comp",
                                start + 2 * m * step,
                                start + 2 * m * step + b, m))
            log.append(LogEvent(stage, f"bwdB.mb{m}", "bwd", "comm",
                                start + (2 * m + 1) * step,
                                start + (2 * m + 1) * step + b, m))
    return log


def export_dualpp_trace(pp: int, mbc: int, f: float, b: float, save_path: str,
                        p2p: float = 0.0, overlap_ratio: float = 1.0):
    from ..sim.trace import events_to_chrome_trace

    log = dualpp_schedule_events(pp, mbc, f, b, p2p, overlap_ratio)
    events_to_chrome_trace(log, save_path)
    return save_path
