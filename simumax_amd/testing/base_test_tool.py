"""Golden-result comparison helpers.

Parity target: simumax/testing/base_test_tool.py:13-83 (ResultCheck,
RelDiffComparator): recursive dict comparison with relative tolerance,
parsing human-readable quantity strings ("66.44 GB", "12.3 ms") back to
numbers.
"""

from __future__ import annotations

from typing import Any, List, Tuple

from ..core.utils import HumanReadableSize


class RelDiffComparator:
    def __init__(self, rel_tol: float = 1e-3, abs_tol: float = 1e-9):
        self.rel_tol = rel_tol
        self.abs_tol = abs_tol

    @staticmethod
    def _to_number(v):
        if isinstance(v, bool):
            return None
        if isinstance(v, (int, float)):
            return float(v)
        if isinstance(v, str):
            try:
                return HumanReadableSize.from_string(v)
            except ValueError:
                return None
        return None

    def equal(self, a, b) -> bool:
        na, nb = self._to_number(a), self._to_number(b)
        if na is not None and nb is not None:
            return abs(na - nb) <= max(self.abs_tol,
                                       self.rel_tol * max(abs(na), abs(nb)))
        return a == b


class ResultCheck:
    """Recursive golden-dict comparison; collects per-path mismatches."""

    def __init__(self, rel_tol: float = 1e-3, ignore_keys=()):
        self.cmp = RelDiffComparator(rel_tol)
        self.ignore_keys = set(ignore_keys)
        self.mismatches: List[Tuple[str, Any, Any]] = []

    def check(self, got, golden, path="") -> bool:
        if isinstance(golden, dict):
            if not isinstance(got, dict):
                self.mismatches.append((path, type(got).__name__, "dict"))
                return False
            ok = True
            for k, gv in golden.items():
                if k in self.ignore_keys:
                    continue
                if k not in got:
                    self.mismatches.append((f"{path}.{k}", "<missing>", gv))
                    ok = False
                    continue
                ok = self.check(got[k], gv, f"{path}.{k}") and ok
            return ok
        if isinstance(golden, (list, tuple)):
            if len(got) != len(golden):
                self.mismatches.append((path, f"len {len(got)}",
                                        f"len {len(golden)}"))
                return False
            ok = True
            for i, (a, b) in enumerate(zip(got, golden)):
                ok = self.check(a, b, f"{path}[{i}]") and ok
            return ok
        if not self.cmp.equal(got, golden):
            self.mismatches.append((path, got, golden))
            return False
        return True

    def report(self) -> str:
        if not self.mismatches:
            return "OK"
        return "\n".join(f"{p}: got {g!r}, expected {e!r}"
                        for p, g, e in self.mismatches[:50])
