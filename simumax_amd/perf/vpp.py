"""Interleaved (sync-VPP) pipeline schedule — exact recurrence.

Parity target: the reference's sync-VPP Preview
(simumax/core/perf_llm.py:1745-1828, 2325-2606 and
pipeline_schedule.py:97-715): Megatron's interleaved 1F1B schedule table
replayed with blocking p2p. Virtual stage v = chunk*pp + stage; each rank
runs vp model chunks; microbatches advance in groups of pp.
"""

from __future__ import annotations

from typing import List

from .perf_llm import ScheduleRecord


def chunk_id_of(k: int, pp: int, vp: int, forward: bool) -> int:
    """Model-chunk index of the k-th fwd (or bwd) on a rank (Megatron
    schedules.get_model_chunk_id with microbatch_group_size = pp)."""
    c = (k // pp) % vp
    return c if forward else (vp - 1 - c)

def mb_id_of(k: int, pp: int, vp: int) -> int:
    """Microbatch index of the k-th fwd (or bwd) on a rank."""
    return (k // (pp * vp)) * pp + k % pp


def schedule_interleaved(pp: int, vp: int, mbc: int,
                         fwd, bwd, p2p: float):
    """fwd/bwd: [stage][chunk] times per microbatch. Returns (total,
    records). Requires mbc % pp == 0 (Megatron constraint)."""
    assert mbc % pp == 0, "interleaved schedule requires micro_batch_num % pp == 0"
    total_per_rank = mbc * vp

    streams = []
    for s in range(pp):
        warm = min((pp - s - 1) * 2 + (vp - 1) * pp, total_per_rank)
        ops = [("F", k) for k in range(warm)]
        nf, nb = warm, 0
        while nb < total_per_rank:
            if nf < total_per_rank:
                ops.append(("F", nf)); nf += 1
            ops.append(("B", nb)); nb += 1
        streams.append(ops)

    # completion times indexed by virtual stage v = c*pp + s and microbatch
    nv = pp * vp
    f_end = [[None] * mbc for _ in range(nv)]
    b_end = [[None] * mbc for _ in range(nv)]
    ptr = [0] * pp
    t = [0.0] * pp
    records: List[ScheduleRecord] = []
    remaining = sum(len(x) for x in streams)
    while remaining:
        progressed = False
        for s in range(pp):
            while ptr[s] < len(streams[s]):
                kind, k = streams[s][ptr[s]]
                if kind == "F":
                    c = chunk_id_of(k, pp, vp, True)
                    m = mb_id_of(k, pp, vp)
                    v = c * pp + s
                    if v == 0:
                        dep = 0.0
                    else:
                        pv = v - 1
                        pe = f_end[pv][m]
                        if pe is None:
                            break
                        dep = pe + p2p
                    dur = fwd[s][c]
                else:
                    c = chunk_id_of(k, pp, vp, False)
                    m = mb_id_of(k, pp, vp)
                    v = c * pp + s
                    if f_end[v][m] is None:
                        break
                    if v == nv - 1:
                        dep = f_end[v][m]
                    else:
                        ne = b_end[v + 1][m]
                        if ne is None:
                            break
                        dep = ne + p2p
                    dur = bwd[s][c]
                start = max(t[s], dep)
                end = start + dur
                records.append(ScheduleRecord(s, m, kind, start, end))
                (f_end if kind == "F" else b_end)[v][m] = end
                t[s] = end
                ptr[s] += 1
                remaining -= 1
                progressed = True
        if not progressed:
            raise RuntimeError("interleaved schedule deadlock")
    return max(t), records


def interleaved_inflight_microbatches(pp: int, vp: int, mbc: int, stage: int) -> int:
    """Max fwd activations in flight on a stage (warmup depth + 1)."""
    warm = min((pp - stage - 1) * 2 + (vp - 1) * pp, mbc * vp)
    return min(mbc * vp, warm + 1)
