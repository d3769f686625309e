"""Render the perf-vs-real validation table as charts (reference parity:
tools/b200/plot_release_charts.py). Reads a validation JSONL (default:
profiles/validation_r02_final.jsonl, one dict per case as written by
scripts/validation_sweep.py) and writes PNGs next to it."""
import json
import os
import sys

import matplotlib
matplotlib.use("Agg")
import matplotlib.pyplot as plt


def main(path="profiles/validation_r02_final.jsonl"):
    rows, seen = [], {}
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            r = json.loads(line)
            if "case" in r and "measured_ms" in r:
                seen[r["case"]] = r          # last write wins (re-runs)
    rows = list(seen.values())
    if not rows:
        print("no rows"); return
    names = [r["case"] for r in rows]
    meas = [r["measured_ms"] for r in rows]
    pred = [r["predicted_ms"] for r in rows]
    terr = [r["timing_err_pct"] for r in rows]
    merr = [r["mem_err_pct"] for r in rows]

    fig, (ax1, ax2) = plt.subplots(
        2, 1, figsize=(11, 8), height_ratios=[2, 1], sharex=True)
    x = range(len(rows))
    w = 0.38
    ax1.bar([i - w / 2 for i in x], meas, w, label="measured ms/step",
            color="#3b6ea5")
    ax1.bar([i + w / 2 for i in x], pred, w, label="predicted ms/step",
            color="#d08a2e")
    ax1.set_ylabel("ms / step")
    ax1.set_title("Predicted vs measured step time — 1x MI355X "
                  "(self-calibrated per box)")
    ax1.legend()
    ax1.grid(axis="y", alpha=0.3)

    ax2.bar([i - w / 2 for i in x], terr, w, label="timing err %",
            color="#3b6ea5")
    ax2.bar([i + w / 2 for i in x], merr, w, label="peak-mem err %",
            color="#9a3b3b")
    ax2.axhline(0, color="black", lw=0.8)
    ax2.set_ylabel("error %")
    ax2.set_xticks(list(x))
    ax2.set_xticklabels(names, rotation=35, ha="right", fontsize=8)
    ax2.legend()
    ax2.grid(axis="y", alpha=0.3)

    fig.tight_layout()
    out = os.path.splitext(path)[0] + ".png"
    fig.savefig(out, dpi=130)
    print(f"wrote {out} ({len(rows)} cases)")


if __name__ == "__main__":
    main(*sys.argv[1:])
