"""CPU tests for the perf-vs-real pipeline's phased state + resume
(reference parity: run_megatron_perf_real_pipeline's state JSONs with
--reuse-state)."""

import json

from simumax_amd.calib import perf_real_pipeline as P


def test_state_roundtrip_and_summary(tmp_path):
    path = str(tmp_path / "state.json")
    state = P._load_state(path)
    assert state == {"cases": {}}
    state["cases"]["a"] = dict(predicted_ms=100.0, predicted_bytes=2.0 * 2**30,
                               measured_ms=104.0, measured_bytes=2.1 * 2**30)
    state["cases"]["incomplete"] = dict(predicted_ms=50.0,
                                        predicted_bytes=1.0)  # no real run yet
    summary = P.summarize(state)
    P._save_state(path, state)

    assert len(summary["rows"]) == 1          # incomplete case excluded
    row = summary["rows"][0]
    assert row["rel_err"] == -3.85            # (100-104)/104
    assert summary["timing_err_range"] == [-3.85, -3.85]

    # resume: reload keeps both phases' data
    again = P._load_state(path)
    assert again["cases"]["incomplete"]["predicted_ms"] == 50.0
    assert again["summary"]["rows"][0]["case"] == "a"


def test_perf_screen_phase_runs_on_cpu():
    out = P.perf_screen(dict(model="llama2-tiny", seq=128, mbs=1, mbc=1))
    assert out["predicted_ms"] > 0
    assert out["predicted_bytes"] > 0
