"""Core utilities: human-readable formatting + rank-group math.

Parity target: simumax/core/utils.py (HumanReadableSize, result formatting,
get_rank_group / get_pp_stage_representative_rank / get_pp_p2p_comm_size).

Note the reference quirk we preserve: analysis_mem results are formatted
STRINGS ("66.44 GB") and the search code parses them back via
HumanReadableSize.from_string.
"""

from __future__ import annotations

import re
from typing import Dict, List


class HumanReadableSize:
    """Format/parse quantities with binary (bytes) or metric (flops) units."""

    BYTE_UNITS = ["B", "KB", "MB", "GB", "TB", "PB"]
    METRIC_UNITS = ["", "K", "M", "G", "T", "P", "E"]

    @staticmethod
    def format_bytes(n: float, precision: int = 2) -> str:
        x = float(n)
        for unit in HumanReadableSize.BYTE_UNITS:
            if abs(x) < 1024.0 or unit == HumanReadableSize.BYTE_UNITS[-1]:
                return f"{x:.{precision}f} {unit}"
            x /= 1024.0
        return f"{x:.{precision}f} PB"

    @staticmethod
    def format_metric(n: float, suffix: str = "", precision: int = 2) -> str:
        x = float(n)
        for unit in HumanReadableSize.METRIC_UNITS:
            if abs(x) < 1000.0 or unit == HumanReadableSize.METRIC_UNITS[-1]:
                return f"{x:.{precision}f} {unit}{suffix}"
            x /= 1000.0
        return f"{x:.{precision}f} E{suffix}"

    @staticmethod
    def format_time_ms(ms: float, precision: int = 2) -> str:
        return f"{float(ms):.{precision}f} ms"

    _PARSE_RE = re.compile(r"^\s*(-?[\d.]+(?:e[+-]?\d+)?)\s*([A-Za-z]*)\s*$")

    @staticmethod
    def from_string(s: str) -> float:
        """Parse '66.44 GB' / '12.3 ms' / '1.5 T' back to base units."""
        if isinstance(s, (int, float)):
            return float(s)
        m = HumanReadableSize._PARSE_RE.match(s)
        if not m:
            raise ValueError(f"cannot parse quantity: {s!r}")
        val = float(m.group(1))
        unit = m.group(2)
        if unit in ("ms", "MS"):
            return val
        if unit.upper().endswith("B") and unit.upper() in HumanReadableSize.BYTE_UNITS:
            return val * 1024.0 ** HumanReadableSize.BYTE_UNITS.index(unit.upper())
        # strip a whole FLOPS/TGS-style suffix — a char-set rstrip would
        # also eat the P of PFLOPS and mis-scale peta quantities by 1e15
        metric = unit
        for suf in ("FLOPS", "FLOPs", "Flops", "flops", "TGS", "tgs"):
            if metric.endswith(suf):
                metric = metric[: -len(suf)]
                break
        if metric.upper() in ("", "K", "M", "G", "T", "P", "E"):
            return val * 1000.0 ** (
                HumanReadableSize.METRIC_UNITS.index(metric.upper())
                if metric.upper() in HumanReadableSize.METRIC_UNITS else 0
            )
        raise ValueError(f"unknown unit in {s!r}")


def human_readable_result(d):
    """Recursively format *_time (ms), *_mem/*_bytes (bytes), *_flops keys."""
    if isinstance(d, list):
        return [human_readable_result(x) for x in d]
    if not isinstance(d, dict):
        return d
    out = {}
    for k, v in d.items():
        if isinstance(v, dict) or isinstance(v, list):
            out[k] = human_readable_result(v)
        elif isinstance(v, (int, float)) and not isinstance(v, bool):
            if k.endswith("_time") or "_time_" in k:
                out[k] = HumanReadableSize.format_time_ms(v)
            elif k.endswith(("_mem", "_bytes", "_cache", "_size_bytes")) or "_mem_" in k:
                out[k] = HumanReadableSize.format_bytes(v)
            elif k.endswith("_flops"):
                out[k] = HumanReadableSize.format_metric(v, "FLOPS")
            else:
                out[k] = v
        else:
            out[k] = v
    return out


# --------------------------------------------------------------------------
# rank <-> parallel-group math (order tp-cp-dp-pp; experts ep-etp-edp-pp)
# --------------------------------------------------------------------------
def get_rank_group(rank: int, strategy) -> Dict[str, int]:
    """Return the coordinates and group ids of `rank` under the canonical
    Megatron order tp-cp-dp-pp (dense) / etp-ep-edp-pp (experts)."""
    tp, cp, pp = strategy.tp_size, strategy.cp_size, strategy.pp_size
    dp = strategy.dp_size
    world = strategy.world_size
    assert 0 <= rank < world
    tp_rank = rank % tp
    cp_rank = (rank // tp) % cp
    dp_rank = (rank // (tp * cp)) % dp
    pp_rank = rank // (tp * cp * dp)

    ep, etp = strategy.ep_size, strategy.etp_size
    edp = strategy.edp_size
    etp_rank = rank % etp
    ep_rank = (rank // etp) % ep
    edp_rank = (rank // (etp * ep)) % edp

    return dict(
        tp_rank=tp_rank, cp_rank=cp_rank, dp_rank=dp_rank, pp_rank=pp_rank,
        etp_rank=etp_rank, ep_rank=ep_rank, edp_rank=edp_rank,
        tp_group=rank // tp,
        cp_group=tp_rank + (rank // (tp * cp)) * tp,
        dp_group=rank % (tp * cp) + pp_rank * tp * cp * dp,
        pp_group=rank % (tp * cp * dp),
        ep_group=etp_rank + (rank // (etp * ep)) * etp,
        edp_group=rank % (etp * ep) + (rank // (etp * ep * edp)) * etp * ep * edp,
    )


def get_pp_stage_representative_rank(stage: int, strategy) -> int:
    """First world rank of PP stage `stage` (lane merging in the simulator)."""
    per_stage = strategy.world_size // strategy.pp_size
    return stage * per_stage


def get_pp_p2p_comm_size(strategy, model_config, dtype_bytes: int = 2) -> int:
    """Bytes of the activation tensor crossing a PP stage boundary per
    microbatch (seq/cp [and /tp under SP] x mbs x hidden)."""
    seq = strategy.seq_len // strategy.cp_size
    if strategy.enable_sequence_parallel:
        seq //= strategy.tp_size
    return seq * strategy.micro_batch_size * model_config.hidden_size * dtype_bytes


def stage_layers(strategy, model_config) -> List[int]:
    """Number of transformer layers per PP stage, honoring the uneven
    first/last-stage overrides and embedding/loss-in-split flags."""
    pp = strategy.pp_size
    total = model_config.layer_num
    first = strategy.num_layers_in_first_pipeline_stage
    last = strategy.num_layers_in_last_pipeline_stage
    extra_slots = int(strategy.account_for_embedding_in_pipeline_split) + int(
        strategy.account_for_loss_in_pipeline_split
    )
    if pp == 1:
        return [total]
    if first is None and last is None and extra_slots == 0:
        assert total % pp == 0, (
            f"layer_num {total} not divisible by pp {pp}; set "
            "num_layers_in_first/last_pipeline_stage"
        )
        return [total // pp] * pp
    if extra_slots and first is None and last is None:
        # Megatron: embedding/loss occupy one layer slot on first/last stage
        padded = total + extra_slots
        assert padded % pp == 0
        per = padded // pp
        layers = [per] * pp
        if strategy.account_for_embedding_in_pipeline_split:
            layers[0] -= 1
        if strategy.account_for_loss_in_pipeline_split:
            layers[-1] -= 1
        return layers
    mid_stages = pp - int(first is not None) - int(last is not None)
    mid_total = total - (first or 0) - (last or 0)
    assert mid_stages >= 0 and mid_total >= 0
    if mid_stages == 0:
        layers = []
    else:
        assert mid_total % mid_stages == 0, (
            f"middle layers {mid_total} not divisible by {mid_stages} stages"
        )
        layers = [mid_total // mid_stages] * mid_stages
    if first is not None:
        layers = [first] + layers
    if last is not None:
        layers = layers + [last]
    return layers
