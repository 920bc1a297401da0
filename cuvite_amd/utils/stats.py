"""Graph distribution statistics (ref printStats, distgraph.hpp:100-149):
per-rank vertex/edge/ghost counts and the edge-count imbalance table."""

from __future__ import annotations

import torch


def print_dist_stats(dg, comm):
    dev = comm.device
    row = torch.tensor(
        [float(dg.nv), float(dg.ne), float(dg.ghost_vertices().numel())],
        dtype=torch.float64, device=dev)
    rows = comm.gather_cat(row, root=0)
    if comm.rank != 0:
        return
    rows = rows.reshape(comm.world, 3).cpu()
    ne = rows[:, 1]
    tot_ne = float(ne.sum())
    mean = tot_ne / comm.world
    print("Graph distribution statistics")
    print(f"{'rank':>5} {'nv':>12} {'ne':>14} {'ghosts':>12} {'ne/mean':>8}")
    for p in range(comm.world):
        print(f"{p:>5} {int(rows[p, 0]):>12} {int(rows[p, 1]):>14} "
              f"{int(rows[p, 2]):>12} {float(rows[p, 1]) / max(mean, 1):>8.3f}")
    print(f"total ne(directed)={int(tot_ne)} "
          f"imbalance(max/mean)={float(ne.max()) / max(mean, 1):.3f}")
