"""Merge fresh measurements from gpurun_out/calib/ into the tracked
calib_raw/ tables. ALWAYS use this instead of cp: every GPU box starts
with an empty gpurun_out, so its calib files hold only that run's keys —
a cp would clobber the accumulated history (this bit us once)."""
import glob
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

for src in sorted(glob.glob("gpurun_out/calib/*.json")):
    dst = os.path.join("calib_raw", os.path.basename(src))
    base = {}
    if os.path.exists(dst):
        with open(dst) as f:
            base = json.load(f)
    with open(src) as f:
        fresh = json.load(f)
    base.update(fresh)
    with open(dst, "w") as f:
        json.dump(base, f, indent=1, sort_keys=True)
    print(f"{dst}: +{len(fresh)} fresh -> {len(base)} total")
