"""CLI smoke tests: every subcommand the README/tutorial advertises must
at least run end to end on CPU with shipped configs."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=240):
    return subprocess.run([sys.executable, "-m", "simumax_amd", *args],
                          capture_output=True, text=True, cwd=REPO,
                          timeout=timeout)


@pytest.mark.timeout(300)
def test_cli_analyze(tmp_path):
    r = _run(["analyze", "--model", "llama2-tiny",
              "--strategy", "tp1_pp1_dp8_mbs1", "--system", "mi355x",
              "--set", "seq_len=256", "--output", str(tmp_path)])
    assert r.returncode == 0, r.stderr[-800:]
    assert "MFU" in r.stdout
    assert (tmp_path / "compute_result.json").exists()
    assert (tmp_path / "mem_result.json").exists()


@pytest.mark.timeout(300)
def test_cli_simulate(tmp_path):
    r = _run(["simulate", "--model", "llama2-tiny",
              "--strategy", "tp1_pp2_dp4_mbs1", "--system", "mi355x",
              "--set", "seq_len=256", "--output", str(tmp_path)])
    assert r.returncode == 0, r.stderr[-800:]
    trace = tmp_path / "tracing_logs.json"
    assert trace.exists(), os.listdir(tmp_path)
    with open(trace) as f:
        t = json.load(f)
    assert t["traceEvents"], "empty chrome trace"
    assert (tmp_path / "simu_memory_snapshot.json").exists()


@pytest.mark.timeout(300)
def test_cli_capture(tmp_path):
    out = tmp_path / "graph"
    r = _run(["capture", "--model", "llama2-tiny",
              "--strategy", "tp1_pp1_dp8_mbs1", "--system", "mi355x",
              "--set", "seq_len=256", "--output", str(out)])
    assert r.returncode == 0, r.stderr[-800:]
    with open(str(out) + ".json") as f:
        g = json.load(f)
    assert g, "empty captured graph"
    assert os.path.exists(str(out) + ".dot")


@pytest.mark.timeout(120)
def test_cli_configs():
    r = _run(["configs"])
    assert r.returncode == 0
    assert "llama3-8b" in r.stdout and "mi355x" in r.stdout
