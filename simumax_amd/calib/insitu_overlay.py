"""Apply an in-situ calibration summary (kernels/insitu.summarize()) to a
live SystemConfig: per-shape operator efficiencies measured on THIS box
override the shipped tables, so predictions track the box the benchmark
actually runs on (same-machine calibrate-then-validate, compressed into
the benchmark process; measured step times vary ±3-5% box-to-box from
clocks alone)."""

from __future__ import annotations


def apply_insitu_overlay(system_config, summary) -> int:
    """Overlay {table: {key: {eff,...}}} rows onto the config. Returns the
    number of overlaid entries."""
    acc = system_config.accelerator
    n = 0
    for table, rows in summary.items():
        if table == "network":
            # measured collective wall time -> rescale the tier op's
            # efficiency so the model's prediction for that (op, bytes,
            # comm_num) matches the measurement exactly
            for op_name, row in rows.items():
                cfg = e0 = None
                try:
                    net = system_config.networks[row.get(
                        "net", "high_intra_node")]
                    cfg = net.op[op_name]
                    args = (op_name, row["bytes"], row["comm_num"])
                    kw = dict(net=row.get("net", "high_intra_node"))
                    # the model is affine in 1/eff (bw term + additive
                    # latency): probe at two eff values, solve exactly
                    e0 = cfg.efficient_factor
                    t1 = system_config.compute_net_op_time(*args, **kw)
                    cfg.efficient_factor = e0 / 2
                    t2 = system_config.compute_net_op_time(*args, **kw)
                    A = (t2 - t1) * e0          # bw term at eff=1
                    L = t1 - A / e0             # additive latency
                    denom = max(row["ms"] - L, 1e-6)
                    cfg.efficient_factor = round(
                        min(max(A / denom, 0.02), 1.5), 4)
                    n += 1
                except (KeyError, AttributeError, TypeError,
                        ZeroDivisionError):
                    if cfg is not None and e0 is not None:
                        cfg.efficient_factor = e0
                    continue
            continue
        if table == "meta":
            if "recompute_factor" in rows:
                acc.recompute_factor = float(rows["recompute_factor"])
                n += 1
            continue
        if table == "bandwidth":
            for k, v in rows.items():
                if k.endswith("_eff"):
                    op = k[:-4]
                    if op == "default":
                        op = "default"
                    bw = acc.bandwidth.get(op)
                    if bw is not None:
                        bw.efficient_factor = round(float(v), 4)
                        n += 1
                elif k == "moe_routing_ms":
                    bw = acc.bandwidth.get("moe_routing")
                    if bw is not None:
                        bw.latency_us = round(float(v) * 1e3, 1)
                        n += 1
                elif k == "moe_routing_bwd_ms":
                    bw = acc.bandwidth.get("moe_routing_bwd")
                    if bw is not None:
                        bw.latency_us = round(float(v) * 1e3, 1)
                        n += 1
                elif k == "optimizer_eff":
                    bw = acc.bandwidth.get("optimizer")
                    if bw is not None:
                        bw.efficient_factor = round(float(v), 4)
                        n += 1
            continue
        op = acc.op.get(table)
        if op is None:
            continue
        aef = op.accurate_efficient_factor
        if aef is None:
            aef = op.accurate_efficient_factor = {}
        for key, row in rows.items():
            if isinstance(row, dict) and "eff" in row:
                aef[key] = round(float(row["eff"]), 4)
                n += 1
    return n
