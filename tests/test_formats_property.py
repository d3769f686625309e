"""Property-based tests (hypothesis) for the format/parse layer the
search path depends on: HumanReadableSize round-trips (analysis_mem
results are STRINGS parsed back during search) and the compact
strategy-string parser."""

from hypothesis import given, settings, strategies as st

from simumax_amd.core.config import StrategyConfig
from simumax_amd.core.utils import HumanReadableSize as H


@given(st.floats(min_value=1e-2, max_value=1e18, allow_nan=False,
                 allow_infinity=False))
@settings(max_examples=200, deadline=None)
def test_bytes_roundtrip(n):
    # 6 printed digits bound the round-trip error (abs for sub-1 values)
    s = H.format_bytes(n, precision=6)
    back = H.from_string(s)
    assert abs(back - n) <= max(1e-4 * n, 1e-6)


@given(st.floats(min_value=1e-2, max_value=1e18, allow_nan=False,
                 allow_infinity=False))
@settings(max_examples=200, deadline=None)
def test_metric_roundtrip(n):
    s = H.format_metric(n, suffix="FLOPS", precision=6)
    back = H.from_string(s)
    assert abs(back - n) <= max(1e-4 * n, 1e-6)


def test_pflops_parse_regression():
    """'746.38 PFLOPS' once parsed to 746.38 — the char-set rstrip ate
    the peta prefix."""
    assert H.from_string("746.384705 PFLOPS") == 746.384705 * 1000.0 ** 5


@given(st.floats(min_value=0, max_value=1e7, allow_nan=False,
                 allow_infinity=False))
@settings(max_examples=50, deadline=None)
def test_time_roundtrip(ms):
    assert abs(H.from_string(H.format_time_ms(ms, 6)) - ms) <= 1e-5 * max(ms, 1)


_pow2 = st.sampled_from([1, 2, 4, 8])


@given(seq=st.sampled_from([2048, 4096, 8192]), mbs=st.sampled_from([1, 2]),
       tp=_pow2, pp=st.sampled_from([1, 2]), cp=st.sampled_from([1, 2]),
       ep=st.sampled_from([1, 2]))
@settings(max_examples=60, deadline=None)
def test_compact_string_roundtrip(seq, mbs, tp, pp, cp, ep):
    """parallelism() emits the compact string; parsing it back must
    reproduce every parallel degree."""
    cfg = StrategyConfig(seq_len=seq, micro_batch_size=mbs,
                         micro_batch_num=2, world_size=tp * pp * cp * ep * 2,
                         tp_size=tp, pp_size=pp, cp_size=cp, ep_size=ep)
    s = cfg.parallelism + f" world_size:{cfg.world_size}"
    back = StrategyConfig.init_from_format_strings(s)
    for f in ("seq_len", "micro_batch_size", "tp_size", "pp_size",
              "cp_size", "ep_size", "world_size"):
        assert getattr(back, f) == getattr(cfg, f), (f, s)
