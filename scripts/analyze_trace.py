"""Bucket a rocprofv3 kernel-trace CSV into op families per step.

Usage: python scripts/analyze_trace.py <kernel_trace.csv> <out.json>
Splits the trace into spans at >50 ms host gaps, takes the last span
(the timed steps), and reports per-bucket busy time and launch counts,
plus the top individual kernels inside the catch-all buckets.
"""
import csv
import json
import sys
from collections import Counter, defaultdict


def bucket(n):
    nl = n.lower()
    if 'cijk' in nl or 'gemm' in nl:
        return 'gemm'
    if 'fa_fwd' in n or 'fa_bwd' in n:
        return 'flash_attn'
    if 'rmsnorm' in n:
        return 'rmsnorm'
    if 'rope' in n:
        return 'rope'
    if 'swiglu' in n:
        return 'swiglu'
    if 'ce_fwd' in n or 'ce_bwd' in n:
        return 'cross_entropy'
    if 'multi_tensor' in n or 'addcdiv' in n:
        return 'optimizer'
    if 'radix' in nl or 'sort' in nl or 'scan' in nl or 'cub' in nl or 'rocprim' in nl:
        return 'sort_scan'
    if 'index' in nl or 'gather' in nl or 'scatter' in nl:
        return 'index_ops'
    if 'copybuffer' in nl or 'copy' in nl or 'catarray' in nl:
        return 'copy'
    if 'fill' in nl:
        return 'fill'
    if 'reduce' in nl:
        return 'reduce'
    if 'elementwise' in nl:
        return 'elementwise'
    return 'other'


def main():
    path, out_path = sys.argv[1], sys.argv[2]
    rows = []
    with open(path) as f:
        for r in csv.DictReader(f):
            rows.append((int(r['Start_Timestamp']), int(r['End_Timestamp']),
                         r['Kernel_Name']))
    rows.sort()
    spans = []
    cur = [rows[0]]
    for r in rows[1:]:
        if r[0] - cur[-1][1] > 50_000_000:
            spans.append(cur)
            cur = []
        cur.append(r)
    spans.append(cur)
    s = max(spans, key=lambda sp: sp[-1][1] - sp[0][0])
    wall = (s[-1][1] - s[0][0]) / 1e6
    busy = defaultdict(float)
    cnt = Counter()
    top = defaultdict(float)
    tc = Counter()
    for a, b, n in s:
        k = bucket(n)
        busy[k] += (b - a) / 1e6
        cnt[k] += 1
        if k in ('other', 'elementwise', 'index_ops', 'copy', 'fill',
                 'sort_scan', 'reduce'):
            key = n.split('<')[0][:90]
            top[key] += (b - a) / 1e6
            tc[key] += 1
    out = dict(
        wall_ms=round(wall, 2),
        n_kernels=len(s),
        buckets={k: dict(ms=round(v, 2), n=cnt[k])
                 for k, v in sorted(busy.items(), key=lambda kv: -kv[1])},
        top_glue=[dict(ms=round(v, 3), n=tc[k], name=k)
                  for k, v in sorted(top.items(), key=lambda kv: -kv[1])[:25]],
        spans=[[len(sp), round((sp[-1][1] - sp[0][0]) / 1e6, 1)]
               for sp in spans],
    )
    with open(out_path, 'w') as f:
        json.dump(out, f, indent=1)
    print(json.dumps(out, indent=1))


if __name__ == '__main__':
    main()
