"""Streamlit GUI over PerfLLM (reference parity: app/streamlit_app.py).

This deployment image has no streamlit; install it to use the GUI, or use
the CLI: `python -m simumax_amd analyze|simulate|search|capture`.
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

try:
    import streamlit as st
except ImportError as e:  # pragma: no cover
    raise SystemExit(
        "streamlit is not installed; use `python -m simumax_amd` instead"
    ) from e

from simumax_amd import (ModelConfig, PerfLLM, StrategyConfig, SystemConfig,
                         get_simu_model_config, get_simu_strategy_config,
                         get_simu_system_config, show_simu_model_configs,
                         show_simu_strategy_configs, show_simu_system_configs)

st.set_page_config(page_title="simumax_amd", layout="wide")
st.title("simumax_amd — MI355X LLM training simulator")

col1, col2, col3 = st.columns(3)
model = col1.selectbox("Model", show_simu_model_configs(), index=0)
strategy = col2.selectbox("Strategy", show_simu_strategy_configs(), index=0)
system = col3.selectbox("System", show_simu_system_configs(), index=0)

with st.sidebar:
    st.header("Overrides")
    tp = st.select_slider("tp_size", [1, 2, 4, 8], 1)
    pp = st.select_slider("pp_size", [1, 2, 4, 8], 1)
    mbs = st.number_input("micro_batch_size", 1, 64, 1)
    mbc = st.number_input("micro_batch_num", 1, 128, 8)
    seq = st.number_input("seq_len", 512, 262144, 4096, step=512)
    recompute = st.selectbox("recompute", ["none", "selective_recompute",
                                           "full_block"])

if st.button("Run analysis"):
    perf = PerfLLM()
    stc = StrategyConfig.init_from_config_file(get_simu_strategy_config(strategy))
    stc.tp_size, stc.pp_size = tp, pp
    stc.micro_batch_size, stc.micro_batch_num, stc.seq_len = mbs, mbc, seq
    stc.enable_recompute = recompute != "none"
    stc.recompute_granularity = None if recompute == "none" else recompute
    perf.configure(
        stc,
        ModelConfig.init_from_config_file(get_simu_model_config(model)),
        SystemConfig.init_from_config_file(get_simu_system_config(system)),
    )
    perf.run_estimate()
    cost = perf.analysis_cost()
    mem = perf.analysis_mem()
    c1, c2, c3, c4 = st.columns(4)
    c1.metric("MFU", f"{cost['mfu']*100:.2f}%")
    c2.metric("iter time", f"{cost['iter_time']:.1f} ms")
    c3.metric("tokens/s/GPU", f"{cost['tgs']:.0f}")
    c4.metric("peak mem", f"{mem['max_peak_mem']/2**30:.1f} GiB")
    st.subheader("Per-stage")
    st.json(mem["stages"])
    st.subheader("Cost breakdown")
    st.json({k: cost[k] for k in ("pipeline_time", "bubble_time", "dp_time",
                                  "optim_time", "straggler_ratio")})
