"""Summarize rocprofv3 PMC csv for fa_ kernels (per-kernel counter sums)."""
import csv, sys, collections, glob
agg = collections.defaultdict(float)
for path in glob.glob(sys.argv[1]):
    with open(path) as f:
        for row in csv.DictReader(f):
            kn = row["Kernel_Name"]
            if "fa_" not in kn:
                continue
            agg[(kn.split("<")[0].split("void ")[-1], row["Counter_Name"])] += \
                float(row["Counter_Value"])
for kname in sorted({k for k, _ in agg}):
    wc = agg.get((kname, "SQ_WAVE_CYCLES"), 0) or 1
    print("==", kname)
    for c in sorted({c for _, c in agg}):
        v = agg.get((kname, c), 0)
        print(f"  {c:28s} {v:.3e}  {v/wc*100:5.1f}%")
