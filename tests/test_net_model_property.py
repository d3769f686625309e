"""Property tests for the RCCL/xGMI network cost model: basic sanity
invariants that must hold for any collective pricing."""

import pytest
from hypothesis import given, settings, strategies as st

from simumax_amd import SystemConfig, get_simu_system_config


@pytest.fixture(scope="module")
def sysc():
    return SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))


OPS = ["all_reduce", "all_gather", "reduce_scatter", "all2all"]


@given(op=st.sampled_from(OPS),
       mb=st.integers(min_value=1, max_value=4096),
       comm=st.sampled_from([2, 4, 8]))
@settings(max_examples=120, deadline=None)
def test_net_time_positive_and_monotonic_in_bytes(op, mb, comm):
    sc = SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))
    small = sc.compute_net_op_time(op, mb * 2**20, comm,
                                   net="high_intra_node")
    big = sc.compute_net_op_time(op, 2 * mb * 2**20, comm,
                                 net="high_intra_node")
    assert small > 0
    assert big >= small


@given(mb=st.sampled_from([64, 256, 1024]))
@settings(max_examples=20, deadline=None)
def test_all_reduce_crossing_nodes_slower(mb):
    """Any intra-node collective is faster than the same bytes priced on
    the inter-node NIC tier."""
    sc = SystemConfig.init_from_config_file(get_simu_system_config("mi355x"))
    intra = sc.compute_net_op_time("all_reduce", mb * 2**20, 8,
                                   net="high_intra_node")
    inter = sc.compute_net_op_time("all_reduce", mb * 2**20, 16,
                                   net="inter_node")
    assert inter > intra


def test_xgmi_point_to_point_is_single_link(sysc):
    """xGMI is point-to-point: a PP hop rides ONE ~153 GB/s link, while a
    full-node collective stripes over all 7 (FC8 scaling) — so p2p is
    SLOWER than an 8-rank all_reduce of the same bytes (the opposite of
    an NVSwitch mental model), and the p2p time must sit near the
    single-link bound."""
    b = 256 * 2**20
    p2p = sysc.compute_net_op_time("p2p", b, 2, net="high_intra_node")
    ar8 = sysc.compute_net_op_time("all_reduce", b, 8,
                                   net="high_intra_node")
    assert p2p > ar8
    single_link_ms = b / (153e9) * 1e3
    assert 0.7 * single_link_ms < p2p < 3.0 * single_link_ms


def test_fc8_more_ranks_use_more_links(sysc):
    """FC8: a 2-rank all_reduce only uses the participating link pair, an
    8-rank one stripes all 7 links — despite moving more ring traffic the
    8-rank collective is faster for the same byte count."""
    b = 256 * 2**20
    t2 = sysc.compute_net_op_time("all_reduce", b, 2, net="high_intra_node")
    t8 = sysc.compute_net_op_time("all_reduce", b, 8, net="high_intra_node")
    assert t8 < t2
