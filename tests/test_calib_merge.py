"""CPU tests for the calibration merge plumbing: RCCL sweep fits fold
into the network tiers (per-op efficiency + per-comm_num overrides),
and the in-situ overlay round-trips through a SystemConfig."""

import json
import os
import shutil

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_merge_rccl_folds_fits(tmp_path, monkeypatch, capsys):
    import simumax_amd.calib.merge_rccl as MR

    # sandbox: copy the real system config + synthetic sweep fits
    (tmp_path / "configs" / "system").mkdir(parents=True)
    (tmp_path / "gpurun_out" / "calib").mkdir(parents=True)
    shutil.copy(os.path.join(REPO, "configs", "system", "mi355x.json"),
                tmp_path / "configs" / "system" / "mi355x.json")
    fits = {
        2: {"all_reduce": {"efficient_factor": 0.61,
                           "fit_latency_ms": 0.012}},
        8: {"all_reduce": {"efficient_factor": 0.74,
                           "fit_latency_ms": 0.018},
            "all_gather": {"efficient_factor": 0.70,
                           "fit_latency_ms": 0.009}},
    }
    for ws, ops in fits.items():
        with open(tmp_path / "gpurun_out" / "calib" / f"rccl_ws{ws}.json",
                  "w") as f:
            json.dump({"ops": ops}, f)
    monkeypatch.setattr(MR, "REPO", str(tmp_path))
    monkeypatch.setattr(MR, "SYSTEM",
                        str(tmp_path / "configs" / "system" / "mi355x.json"))
    MR.main()

    with open(tmp_path / "configs" / "system" / "mi355x.json") as f:
        sysc = json.load(f)
    op = sysc["networks"]["high_intra_node"]["op"]["all_reduce"]
    # headline eff comes from the largest world size
    assert op["efficient_factor"] == 0.74
    assert op["efficient_factor_by_comm_num"] == {"2": 0.61, "8": 0.74}
    assert op["fixed_latency_us_by_comm_num"] == {"2": 12.0, "8": 18.0}
    # single-ws op gets no by_comm_num tables
    ag = sysc["networks"]["high_intra_node"]["op"]["all_gather"]
    assert ag["efficient_factor"] == 0.70
    assert "efficient_factor_by_comm_num" not in ag

    # the folded config still loads and prices a collective
    from simumax_amd import SystemConfig
    sc = SystemConfig.init_from_config_file(
        str(tmp_path / "configs" / "system" / "mi355x.json"))
    t = sc.compute_net_op_time("all_reduce", 64 * 2**20, 8,
                               net="high_intra_node")
    assert t > 0


def test_merge_rccl_no_files_is_noop(tmp_path, monkeypatch, capsys):
    import simumax_amd.calib.merge_rccl as MR

    monkeypatch.setattr(MR, "REPO", str(tmp_path))
    monkeypatch.setattr(MR, "SYSTEM", str(tmp_path / "nope.json"))
    MR.main()   # must not raise or write
    assert not os.path.exists(tmp_path / "nope.json")
    assert "no rccl_ws" in capsys.readouterr().out
