"""Web front-end over PerfLLM (FastAPI + uvicorn).

The reference ships a Streamlit GUI (app/streamlit_app.py:26-862: pick
model / hardware / parallelism, run the analysis, download the artifact
zip). Streamlit is not in this image, so the MI355X build serves the
same workflow over FastAPI (installed):

    python -m simumax_amd.webapp          # http://127.0.0.1:8642
    python -m uvicorn simumax_amd.webapp:app --port 8642

Endpoints:
    GET  /                  HTML form (model / system / strategy knobs)
    POST /analyze           run PerfLLM analysis -> HTML report
    GET  /api/analyze       same as JSON (query params)
    GET  /api/configs       registered model/strategy/system configs
    GET  /api/artifacts.zip analysis + per-op cost dump as a zip
"""

from __future__ import annotations

import io
import json
import zipfile
from copy import deepcopy

from fastapi import FastAPI, Query, Request
from fastapi.responses import HTMLResponse, Response

from .core.config import ModelConfig, StrategyConfig, SystemConfig
from .perf.perf_llm import PerfLLM
from .registry import (get_simu_model_config, get_simu_strategy_config,
                       get_simu_system_config, show_simu_model_configs,
                       show_simu_strategy_configs, show_simu_system_configs)

app = FastAPI(title="simumax_amd", docs_url="/docs")

_KNOBS = ("world_size", "tp_size", "pp_size", "ep_size", "cp_size",
          "micro_batch_size", "micro_batch_num", "seq_len",
          "interleaving_size")


_STR_KNOBS = ("cp_comm_type", "cp_sharding")


def _run(model: str, system: str, strategy: str, overrides: dict):
    st = StrategyConfig.init_from_config_file(get_simu_strategy_config(strategy))
    for k, v in overrides.items():
        if v is not None:
            setattr(st, k, str(v) if k in _STR_KNOBS else int(v))
    mc = ModelConfig.init_from_config_file(get_simu_model_config(model))
    sysc = SystemConfig.init_from_config_file(get_simu_system_config(system))
    perf = PerfLLM()
    perf.configure(st, deepcopy(mc), sysc)
    perf.run_estimate()
    cost = perf.analysis_cost()
    mem = perf.analysis_mem()
    return perf, dict(
        model=model, system=system, strategy=strategy,
        parallelism=perf.strategy.parallelism,
        iter_time_ms=round(cost["iter_time"], 3),
        mfu=round(cost["mfu"], 4),
        tgs=round(cost["tgs"], 1),
        bubble_time_ms=round(cost["bubble_time"], 3),
        dp_time_ms=round(cost["dp_time"], 3),
        optim_time_ms=round(cost["optim_time"], 3),
        max_peak_mem_gib=round(mem["max_peak_mem"] / 2**30, 3),
        mem_stages=[
            {k: (round(v / 2**30, 3) if isinstance(v, float) else v)
             for k, v in s.items() if not isinstance(v, (dict, list))}
            for s in mem["stages"]
        ],
    )


@app.get("/api/configs")
def api_configs():
    return dict(models=show_simu_model_configs(),
                strategies=show_simu_strategy_configs(),
                systems=show_simu_system_configs())


@app.get("/api/analyze")
def api_analyze(model: str = Query("llama3-8b"),
                system: str = Query("mi355x"),
                strategy: str = Query("tp1_pp1_dp8_mbs1"),
                world_size: int = None, tp_size: int = None,
                pp_size: int = None, ep_size: int = None,
                cp_size: int = None, micro_batch_size: int = None,
                micro_batch_num: int = None, seq_len: int = None,
                interleaving_size: int = None,
                cp_comm_type: str = None, cp_sharding: str = None):
    loc = locals()
    over = {k: loc[k] for k in _KNOBS}
    over.update({k: loc[k] for k in _STR_KNOBS})
    _, res = _run(model, system, strategy, over)
    return res


@app.get("/api/artifacts.zip")
def api_artifacts(model: str = "llama3-8b", system: str = "mi355x",
                  strategy: str = "tp1_pp1_dp8_mbs1"):
    perf, res = _run(model, system, strategy, {})
    buf = io.BytesIO()
    with zipfile.ZipFile(buf, "w", zipfile.ZIP_DEFLATED) as z:
        z.writestr("analysis.json", json.dumps(res, indent=1))
        z.writestr("gemm_cost.json", json.dumps(
            perf.chunks[0].get_all_gemm_cost_info(), indent=1))
        z.writestr("op_info.json", json.dumps(
            perf.chunks[0].analysis_op_info(), indent=1, default=str))
    return Response(buf.getvalue(), media_type="application/zip",
                    headers={"Content-Disposition":
                             "attachment; filename=simumax_amd_artifacts.zip"})


def _options(names, selected):
    return "".join(
        f'<option value="{n}"{" selected" if n == selected else ""}>{n}</option>'
        for n in names)


_PAGE = """<!doctype html><html><head><title>simumax_amd</title>
<style>body{{font-family:sans-serif;margin:2em;max-width:60em}}
table{{border-collapse:collapse}}td,th{{border:1px solid #999;padding:4px 10px}}
input,select{{margin:2px}}</style></head><body>
<h2>simumax_amd &mdash; MI355X LLM-training simulator</h2>
<form method="post" action="/analyze">
model <select name="model">{models}</select>
system <select name="system">{systems}</select>
strategy <select name="strategy">{strategies}</select><br>
{fields}
<button type="submit">Analyze</button></form>
{report}
<p><a href="/docs">API docs</a> &middot;
<a href="/api/artifacts.zip">artifacts.zip</a></p></body></html>"""


def _page(report="", model="llama3-8b", system="mi355x",
          strategy="tp1_pp1_dp8_mbs1"):
    fields = "".join(
        f'{k} <input name="{k}" size="4" placeholder="cfg">' + ("<br>" if i % 4 == 3 else "")
        for i, k in enumerate(_KNOBS))
    return _PAGE.format(models=_options(show_simu_model_configs(), model),
                        systems=_options(show_simu_system_configs(), system),
                        strategies=_options(show_simu_strategy_configs(),
                                            strategy),
                        fields=fields, report=report)


@app.get("/", response_class=HTMLResponse)
def index():
    return _page()


@app.post("/analyze", response_class=HTMLResponse)
async def analyze(request: Request):
    # parse urlencoded body directly (python-multipart is not installed)
    from urllib.parse import parse_qs

    body = (await request.body()).decode()
    form = {k: v[0] for k, v in parse_qs(body).items()}
    model = form.get("model", "llama3-8b")
    system = form.get("system", "mi355x")
    strategy = form.get("strategy", "tp1_pp1_dp8_mbs1")
    overrides = {k: (form.get(k) or None) for k in _KNOBS}
    try:
        _, res = _run(model, system, strategy, overrides)
    except (AssertionError, ValueError, KeyError) as e:
        return _page(f"<p style='color:red'>error: {e}</p>", model, system,
                     strategy)
    rows = "".join(f"<tr><th>{k}</th><td>{v}</td></tr>"
                   for k, v in res.items() if not isinstance(v, list))
    report = f"<h3>result</h3><table>{rows}</table>"
    return _page(report, model, system, strategy)


def main():
    import uvicorn

    uvicorn.run(app, host="127.0.0.1", port=8642)


if __name__ == "__main__":
    main()
