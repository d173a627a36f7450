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
