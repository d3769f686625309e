"""Named config registry (parity: simumax/utils.py:26-97)."""

import os

_ROOT = os.path.normpath(os.path.join(os.path.dirname(__file__), ".."))
CONFIG_ROOT = os.path.join(_ROOT, "configs")


def _lookup(kind: str, name: str) -> str:
    path = os.path.join(CONFIG_ROOT, kind, f"{name}.json")
    if not os.path.exists(path):
        avail = _list(kind)
        raise FileNotFoundError(f"{kind} config '{name}' not found; available: {avail}")
    return path


def _list(kind: str):
    d = os.path.join(CONFIG_ROOT, kind)
    if not os.path.isdir(d):
        return []
    return sorted(os.path.splitext(f)[0] for f in os.listdir(d) if f.endswith(".json"))


def get_simu_model_config(name: str) -> str:
    return _lookup("models", name)


def get_simu_strategy_config(name: str) -> str:
    return _lookup("strategy", name)


def get_simu_system_config(name: str) -> str:
    return _lookup("system", name)


def show_simu_model_configs():
    return _list("models")


def show_simu_strategy_configs():
    return _list("strategy")


def show_simu_system_configs():
    return _list("system")
