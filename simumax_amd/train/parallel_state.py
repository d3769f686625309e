"""Process-group grid for composed tp x pp training.

Rank layout follows core/utils.get_rank_group: tp fastest, then dp,
then pp outermost — rank = tp_idx + dp_idx*tp + stage*tp*dp.
(Single-axis tp / ep grids are built by train.tp.get_tp_groups and
train.trainer.get_ep_groups; this module serves the composed case.)

All ranks must call init_parallel_state with the same sizes (new_group
is collective); results are cached per (tp, pp, world).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch.distributed as dist

_CACHE = {}


@dataclass
class ParallelState:
    tp_size: int = 1
    pp_size: int = 1
    dp_size: int = 1
    tp_rank: int = 0
    stage: int = 0
    dp_rank: int = 0
    tp_group: Optional[object] = None
    dp_group: Optional[object] = None        # dense-grad reduction
    pp_prev: Optional[int] = None
    pp_next: Optional[int] = None
    pp_norm_group: Optional[object] = None   # one rank per stage


def init_parallel_state(tp_size=1, pp_size=1) -> ParallelState:
    if not dist.is_initialized() or (tp_size <= 1 and pp_size <= 1):
        return ParallelState(tp_size=tp_size, pp_size=pp_size)
    world = dist.get_world_size()
    assert world % (tp_size * pp_size) == 0
    dp = world // (tp_size * pp_size)
    stage_span = tp_size * dp
    key = (tp_size, pp_size, world)
    if key not in _CACHE:
        grids = dict(tp={}, dp={}, pp_norm={})
        for start in range(0, world, tp_size):
            g = dist.new_group(list(range(start, start + tp_size)))
            for r in range(start, start + tp_size):
                grids["tp"][r] = g
        for stage in range(pp_size):
            base = stage * stage_span
            for off in range(tp_size):
                ranks = list(range(base + off, base + stage_span, tp_size))
                g = dist.new_group(ranks)
                for r in ranks:
                    grids["dp"][r] = g
        for off in range(stage_span):
            ranks = list(range(off, world, stage_span))
            g = dist.new_group(ranks)
            for r in ranks:
                grids["pp_norm"][r] = g
        _CACHE[key] = grids
    grids = _CACHE[key]
    r = dist.get_rank()
    stage = r // stage_span
    within = r % stage_span
    ps = ParallelState(
        tp_size=tp_size, pp_size=pp_size, dp_size=dp,
        tp_rank=within % tp_size, stage=stage, dp_rank=within // tp_size,
        tp_group=grids["tp"][r] if tp_size > 1 else None,
        dp_group=grids["dp"][r],
        pp_norm_group=grids["pp_norm"][r] if pp_size > 1 else None)
    if pp_size > 1:
        ps.pp_prev = r - stage_span if stage > 0 else None
        ps.pp_next = r + stage_span if stage < pp_size - 1 else None
    return ps
