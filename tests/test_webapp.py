"""Web front-end smoke tests (FastAPI TestClient, CPU-only)."""

import io
import zipfile

import pytest

pytest.importorskip("fastapi")

from fastapi.testclient import TestClient  # noqa: E402

from simumax_amd.webapp import app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    return TestClient(app)


def test_index(client):
    r = client.get("/")
    assert r.status_code == 200
    assert "simumax_amd" in r.text


def test_api_configs(client):
    r = client.get("/api/configs")
    assert r.status_code == 200
    d = r.json()
    assert "llama3-8b" in d["models"]
    assert "mi355x" in d["systems"]


def test_api_analyze(client):
    r = client.get("/api/analyze", params=dict(
        model="llama3-8b", system="mi355x", strategy="tp1_pp2_dp4_mbs1"))
    assert r.status_code == 200
    d = r.json()
    assert d["iter_time_ms"] > 0
    assert 0 < d["mfu"] < 1
    assert d["max_peak_mem_gib"] > 0


def test_analyze_form_override(client):
    r = client.post("/analyze", data=dict(
        model="llama3-8b", system="mi355x", strategy="tp1_pp1_dp8_mbs1",
        world_size="8", tp_size="2", pp_size="1", ep_size="", cp_size="",
        micro_batch_size="", micro_batch_num="", seq_len="",
        interleaving_size=""))
    assert r.status_code == 200
    assert "tp2" in r.text


def test_artifacts_zip(client):
    r = client.get("/api/artifacts.zip")
    assert r.status_code == 200
    z = zipfile.ZipFile(io.BytesIO(r.content))
    names = z.namelist()
    assert "analysis.json" in names and "gemm_cost.json" in names


def test_api_analyze_cp_mode_knobs(client):
    r = client.get("/api/analyze", params=dict(
        model="llama2-tiny", strategy="tp1_pp1_dp8_mbs1", system="mi355x",
        seq_len=1024, cp_size=4, cp_comm_type="ring", cp_sharding="zigzag"))
    assert r.status_code == 200
    body = r.json()
    assert body["iter_time_ms"] > 0
    assert "cp4" in body["parallelism"]
