from .cost_model import (all_gather_cost, all_reduce_cost, all_to_all_cost,
                         reduce_scatter_cost, reshard_cost)
from .solver import AutoFlowSolver1D, solve_mesh_dim

__all__ = [
    "AutoFlowSolver1D", "solve_mesh_dim", "reshard_cost",
    "all_gather_cost", "all_reduce_cost", "reduce_scatter_cost",
    "all_to_all_cost",
]
