"""Fold RCCL sweep fits (gpurun_out/calib/rccl_ws{2,4,8}.json) into the
network section of configs/system/mi355x.json: per-op efficient_factor
(from ws=num_per_node) + per-comm_num efficiency/latency overrides.
Parity: one_click_common.py:update_system_network_from_fit +
apply_ws_comm_model.py."""

import glob
import json
import os
import re

REPO = os.path.normpath(os.path.join(os.path.dirname(__file__), "..", ".."))
SYSTEM = os.path.join(REPO, "configs", "system", "mi355x.json")


def main():
    fits = {}
    for p in glob.glob(os.path.join(REPO, "gpurun_out", "calib",
                                    "rccl_ws*.json")):
        ws = int(re.search(r"ws(\d+)", p).group(1))
        with open(p) as f:
            fits[ws] = json.load(f)["ops"]
    if not fits:
        print("no rccl_ws*.json found; run calib.rccl_sweep first")
        return
    with open(SYSTEM) as f:
        sysc = json.load(f)
    net = sysc["networks"]["high_intra_node"]
    full = max(fits)
    for op, fit in fits[full].items():
        cfg = net["op"].setdefault(op, {})
        cfg["efficient_factor"] = round(fit["efficient_factor"], 4)
        by_n = {}
        lat_by_n = {}
        for ws, ops in sorted(fits.items()):
            if op in ops:
                by_n[str(ws)] = round(ops[op]["efficient_factor"], 4)
                lat_by_n[str(ws)] = round(ops[op]["fit_latency_ms"] * 1e3, 2)
        if len(by_n) > 1:
            cfg["efficient_factor_by_comm_num"] = by_n
            cfg["fixed_latency_us_by_comm_num"] = lat_by_n
        print(f"{op}: eff {cfg['efficient_factor']} by_n {by_n}")
    with open(SYSTEM, "w") as f:
        json.dump(sysc, f, indent=1)
    print(f"wrote {SYSTEM}")


if __name__ == "__main__":
    main()
