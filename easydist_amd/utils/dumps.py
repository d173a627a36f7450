"""Debug dump helpers (observability parity with the reference's
DUMP_FX_GRAPH / strategy / MetaIR dumps — SURVEY.md §5).

Set EASYDIST_DUMP_DIR to activate; each compile writes:
  <dir>/<tag>_graph.txt        readable fx graph
  <dir>/<tag>_strategies.txt   solver-chosen per-node strategies
"""
from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

logger = logging.getLogger(__name__)


def dump_dir() -> Optional[str]:
    d = os.environ.get("EASYDIST_DUMP_DIR")
    if d:
        os.makedirs(d, exist_ok=True)
    return d


def dump_graph(gm, tag: str):
    d = dump_dir()
    if not d:
        return
    path = os.path.join(d, f"{tag}_graph.txt")
    with open(path, "w") as f:
        f.write(str(gm.graph))
    logger.info("dumped fx graph to %s", path)


def dump_graph_dot(gm, tag: str):
    """Graphviz .dot text of the fx graph (reference observability:
    DUMP_FX_GRAPH pdf/dot via pygraphviz, compile_auto.py:487-508 —
    plain .dot here, no graphviz dependency; render offline)."""
    d = dump_dir()
    if not d:
        return
    path = os.path.join(d, f"{tag}_graph.dot")
    with open(path, "w") as f:
        f.write("digraph G {\n  rankdir=TB;\n  node [shape=box, "
                "fontsize=9];\n")
        for n in gm.graph.nodes:
            label = f"{n.name}\\n{getattr(n.target, '__name__', n.target)}" \
                if n.op == "call_function" else f"{n.name}\\n[{n.op}]"
            color = {"placeholder": "lightblue", "output": "salmon"}.get(
                n.op, "white")
            f.write(f'  "{n.name}" [label="{label}", style=filled, '
                    f'fillcolor={color}];\n')
            for inp in n.all_input_nodes:
                f.write(f'  "{inp.name}" -> "{n.name}";\n')
        f.write("}\n")
    logger.info("dumped fx graph dot to %s", path)


def dump_strategies(strategies_per_dim: List[Dict], tag: str):
    d = dump_dir()
    if not d:
        return
    path = os.path.join(d, f"{tag}_strategies.txt")
    with open(path, "w") as f:
        for dim, strat in enumerate(strategies_per_dim):
            f.write(f"== mesh dim {dim} ==\n")
            for name, s in sorted(strat.items()):
                f.write(f"{name}: {s}\n")
    logger.info("dumped strategies to %s", path)
