"""Vite-compatible binary graph I/O.

File layout (reference: loadDistGraphMPIIO, distgraph.cpp:69-203 and
writeGraph, distgraph.cpp:936-1014):

    int64  nv
    int64  ne            (directed edge count: each undirected edge twice)
    int64  index[nv+1]   (global CSR offsets, index[0] = 0)
    Edge   edges[ne]     (struct { int64 tail; float64 weight; } = 16 bytes)

Each rank reads only its slice with plain POSIX preads (single-node design;
the reference used MPI-IO collectives, which buy nothing on one node).
"""

from __future__ import annotations

import os
import struct

import numpy as np
import torch

from .graph import Graph, DistGraph, Partition

_HDR = struct.Struct("<qq")  # nv, ne
EDGE_BYTES = 16  # int64 tail + float64 weight


def read_header(path: str):
    with open(path, "rb") as f:
        nv, ne = _HDR.unpack(f.read(_HDR.size))
    return nv, ne


def _read_index_slice(f, base: int, count: int) -> np.ndarray:
    """Read index[base : base+count] (int64 each) at header offset 16."""
    f.seek(16 + 8 * base)
    return np.fromfile(f, dtype="<i8", count=count)


def load_dist_graph(path: str, rank: int, nranks: int,
                    balanced: bool = False,
                    weight_dtype: torch.dtype = torch.float64,
                    unit_weights: bool = False) -> DistGraph:
    """Load this rank's shard of a Vite-format binary graph.

    balanced=True reproduces the -b edge-balanced partition
    (loadDistGraphMPIIOBalanced + balanceEdges, distgraph.cpp:22-66,206-337).
    unit_weights=True replaces all weights with 1.0
    (SET_EDGE_WEIGHTS_TO_ONE, distgraph.cpp:200-202).
    """
    nv, ne = read_header(path)
    with open(path, "rb") as f:
        if balanced:
            index_all = _read_index_slice(f, 0, nv + 1)
            part = Partition.edge_balanced(torch.from_numpy(index_all.copy()), nranks)
            base, bound = part.base(rank), part.bound(rank)
            idx = torch.from_numpy(index_all[base:bound + 1].copy())
        else:
            part = Partition.contiguous(nv, nranks)
            base, bound = part.base(rank), part.bound(rank)
            idx_np = _read_index_slice(f, base, bound - base + 1)
            idx = torch.from_numpy(idx_np.copy())

        e0, e1 = int(idx[0]), int(idx[-1])
        ne_local = e1 - e0
        edge_off = 16 + 8 * (nv + 1) + EDGE_BYTES * e0
        f.seek(edge_off)
        raw = np.fromfile(f, dtype=[("tail", "<i8"), ("weight", "<f8")], count=ne_local)

    rowptr = idx - e0
    tails = torch.from_numpy(raw["tail"].copy())
    if unit_weights:
        weights = torch.ones(ne_local, dtype=weight_dtype)
    else:
        weights = torch.from_numpy(raw["weight"].copy()).to(weight_dtype)
    return DistGraph(Graph(rowptr, tails, weights), part, rank)


def load_graph(path: str, weight_dtype: torch.dtype = torch.float64) -> Graph:
    dg = load_dist_graph(path, 0, 1, weight_dtype=weight_dtype)
    return dg.g


def write_graph(path: str, g: Graph):
    """Write a whole graph in Vite binary format (single writer).

    Reference analog: writeGraph (distgraph.cpp:936-1014); rank shards are
    gathered by the caller before writing (cold path).
    """
    nv, ne = g.nv, g.ne
    with open(path, "wb") as f:
        f.write(_HDR.pack(nv, ne))
        g.rowptr.cpu().numpy().astype("<i8").tofile(f)
        raw = np.empty(ne, dtype=[("tail", "<i8"), ("weight", "<f8")])
        raw["tail"] = g.tails.cpu().numpy()
        raw["weight"] = g.weights.cpu().to(torch.float64).numpy()
        raw.tofile(f)


def write_dist_graph(path: str, shards):
    """Concatenate per-rank shards (list of DistGraph ordered by rank) and write."""
    rowptr = [shards[0].g.rowptr]
    off = int(shards[0].g.rowptr[-1])
    for s in shards[1:]:
        rowptr.append(s.g.rowptr[1:] + off)
        off += int(s.g.rowptr[-1])
    g = Graph(
        torch.cat(rowptr),
        torch.cat([s.g.tails for s in shards]),
        torch.cat([s.g.weights for s in shards]),
    )
    write_graph(path, g)


def load_ground_truth(path: str, zero_based: bool = True) -> torch.Tensor:
    """Load a ground-truth/community file. Two formats are accepted:
    `<vertex> <community>` pairs per line (the LFR ground-truth layout, ref
    loadGroundTruthFile louvain.cpp:3272-3304) or one community id per line
    indexed by line number (the reference's own dump format,
    main.cpp:521-550). Returns community[v]."""
    data = np.loadtxt(path, dtype=np.int64, ndmin=2)
    if data.shape[1] >= 2:
        v, c = data[:, 0], data[:, 1]
        if not zero_based:
            v = v - 1
            c = c - 1
        out = np.zeros(int(v.max()) + 1 if v.size else 0, dtype=np.int64)
        out[v] = c
        return torch.from_numpy(out)
    c = data[:, 0]
    if not zero_based:
        c = c - 1
    return torch.from_numpy(c.copy())


def write_communities(path: str, comm: torch.Tensor):
    """Dump one community id per line, line number = vertex id (exactly the
    reference's <input>.communities format, main.cpp:536-538)."""
    c = comm.cpu().numpy()
    with open(path, "w") as f:
        for v in range(len(c)):
            f.write(f"{int(c[v])}\n")
