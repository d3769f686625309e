"""Operator-graph capture + Graphviz export.

Parity target: simumax/core/graph.py:132-352 (SimuONNXGraphBuilder,
visualize_with_graphviz; capture hook base_struct.py:966-974 and
PerfBase.capture perf_llm.py:476-487): an ONNX-style record of every leaf
op executed during shape propagation, with shapes, op kinds, FLOPs and
the recompute/variance markers, exported as JSON and .dot text.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class GraphNode:
    idx: int
    name: str
    op_type: str
    inputs: List[str]
    outputs: List[str]
    fwd_flops: float = 0.0
    recompute: bool = False
    is_variance_node: bool = False


class SimuGraphBuilder:
    """Collects leaf ops in execution order (attach via
    PathDebugContext.graph_builder before run_estimate)."""

    def __init__(self):
        self.nodes: List[GraphNode] = []

    def add_node(self, leaf):
        self.nodes.append(GraphNode(
            idx=len(self.nodes),
            name=leaf.full_name,
            op_type=type(leaf).__name__,
            inputs=[repr(t) for t in leaf.input_info.tensors],
            outputs=[repr(t) for t in leaf.output_info.tensors],
            fwd_flops=leaf.get_compute_info().fwd_flops if leaf._info_ready
            else leaf._compute_info.fwd_flops,
            recompute=leaf.enable_recompute,
            is_variance_node=leaf.is_variance_node,
        ))

    def mark_recompute_from_modules(self, root):
        """Refresh recompute/variance flags after apply_recompute()."""
        flags = {l.full_name: (l.enable_recompute, l.is_variance_node)
                 for l in root.leaf_modules()}
        for n in self.nodes:
            if n.name in flags:
                n.recompute, n.is_variance_node = flags[n.name]

    def to_dict(self) -> dict:
        return {"nodes": [vars(n) for n in self.nodes]}

    def save_json(self, path: str):
        with open(path, "w") as f:
            json.dump(self.to_dict(), f, indent=1)

    def to_dot(self) -> str:
        """Graphviz digraph (sequential data flow, recompute dashed,
        variance nodes grey)."""
        lines = ["digraph simumax {", "  rankdir=TB;",
                 '  node [shape=box, fontsize=10];']
        for n in self.nodes:
            style = []
            if n.recompute:
                style.append("style=dashed")
            if n.is_variance_node:
                style.append('fillcolor=lightgrey, style="filled,dashed"')
            attr = (", " + ", ".join(style)) if style else ""
            label = f"{n.name}\\n{n.op_type}\\n{n.outputs[0] if n.outputs else ''}"
            lines.append(f'  n{n.idx} [label="{label}"{attr}];')
        for a, b in zip(self.nodes, self.nodes[1:]):
            lines.append(f"  n{a.idx} -> n{b.idx};")
        lines.append("}")
        return "\n".join(lines)

    def save_dot(self, path: str):
        with open(path, "w") as f:
            f.write(self.to_dot())


def capture_graph(perf_model, save_prefix: Optional[str] = None) -> SimuGraphBuilder:
    """Re-run shape propagation with graph capture on (PerfBase.capture
    analog). Call after configure()."""
    from ..core.records import PathDebugContext

    builder = SimuGraphBuilder()
    perf_model.debug_ctx = PathDebugContext(graph_builder=builder)
    perf_model.model_config.maybe_pad_vocab_size(perf_model.strategy.tp_size)
    perf_model.analysis_net(re_analysis=True)
    perf_model.build()
    for stage, chunk in enumerate(perf_model.chunks):
        chunk(perf_model._input_info_for_stage(stage), perf_model.debug_ctx)
        chunk.apply_recompute()
        builder.mark_recompute_from_modules(chunk)
    perf_model._estimated = True
    if save_prefix:
        builder.save_json(save_prefix + ".json")
        builder.save_dot(save_prefix + ".dot")
    return builder
