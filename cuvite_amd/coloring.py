"""Distance-1 coloring by multi-hash min-max (Luby-style).

Faithful reimplementation of distColoringMultiHashMinMax
(coloring.cpp:3-72) and distColoringIteration (coloring.cpp:87-202),
vectorized over vertices/edges; ghost colored-state exchange replaces
sendColoredRemoteVertices (coloring.cpp:320-420).

Per round (seed evolves seed = hash(seed, 0), colors base += 2*nHash):
a still-uncolored vertex v takes color base+2t if hash_t(v) is the strict
minimum over its competing neighbors (uncolored at round start), base+2t+1
if the strict maximum; among its available (t, min/max) slots it picks slot
number (gid % n_available) — exactly the reference's selection. The loop
stops when >= 70% of vertices are colored (MAX_COVG, main.cpp:26) or a round
makes no progress; the leftover vertices form one extra final color class
(louvain.cpp:821,837).
"""

from __future__ import annotations

from typing import Tuple

import torch

from .graph import DistGraph
from .halo import build_halo, exchange_ghost_labels
from .parallel import Comm

MAX_COVG = 70  # percent, ref main.cpp:26


def _hash(a: torch.Tensor, seed: int) -> torch.Tensor:
    """Reference 32-bit mix (coloring.cpp:74-85), vectorized on uint32
    semantics emulated with int64 masking."""
    M = 0xFFFFFFFF
    a = (a & M) ^ (seed & M)
    a = ((a + 0x7ED55D16) + (a << 12)) & M
    a = ((a ^ 0xC761C23C) + (a >> 19)) & M
    a = ((a + 0x165667B1) + (a << 5)) & M
    a = ((a ^ 0xD3A2646C) + (a << 9)) & M
    a = ((a + 0xFD7046C5) + (a << 3)) & M
    a = ((a ^ 0xB55A4F09) + (a >> 16)) & M
    return a


def _hash_scalar(a: int, seed: int) -> int:
    return int(_hash(torch.tensor([a], dtype=torch.int64), seed)[0])


def distance1_coloring(dg: DistGraph, comm: Comm, n_hash: int = 4,
                       halo=None) -> Tuple[torch.Tensor, int]:
    """Returns (colors int64 [nv] in [0, num_colors), num_colors).
    Uncolored leftovers are assigned the last class num_colors-1.
    `halo`: optional prebuilt HaloContext — the phase loop passes the one
    PhaseState will reuse, so -c/-d runs do ghost discovery once per phase
    instead of twice (round-1 weak item 5)."""
    dev = dg.g.device
    nv = dg.nv
    base = dg.base
    if halo is None:
        halo = build_halo(dg, comm)
    rowptr = dg.g.rowptr
    ne = dg.g.ne
    gid_all = torch.cat([torch.arange(base, dg.bound, device=dev), halo.ghosts])
    tnv = dg.nv_global
    # edges are processed in chunks: boolean compaction over a >INT_MAX edge
    # array overflows torch's internal indexing (hit at R-MAT s26), and
    # chunking also avoids materializing 17 GB seg/tail_gid arrays
    CH = 1 << 28

    colors = torch.full((nv,), -1, dtype=torch.int64, device=dev)
    seed = 1012
    next_color = 0
    last_count = 0
    target = (tnv * MAX_COVG) // 100

    use_hip = dev.type == "cuda"
    if use_hip:
        from . import ops
        use_hip = ops.available()

    while True:
        # competing = uncolored at round start (local + ghosts)
        ghost_colors = exchange_ghost_labels(halo, colors)
        uncolored_all = torch.cat([colors, ghost_colors]) < 0
        # strict min / strict max of each hash over competing neighbors
        if use_hip:
            # one wave per vertex, all hashes in a single edge pass
            from . import ops
            seeds = torch.tensor(
                [(seed + 1043 * t) & 0xFFFFFFFF for t in range(n_hash)],
                dtype=torch.int64, device=dev)
            mn2, mx2 = ops._require().coloring_minmax(
                rowptr, halo.tails_dense, gid_all, uncolored_all, colors,
                base, seeds)
            mns = [mn2[:, t] for t in range(n_hash)]
            mxs = [mx2[:, t] for t in range(n_hash)]
        else:
            # chunked torch path (boolean compaction over >INT_MAX edges
            # overflows torch's internal indexing, hence CH slices)
            mns = [torch.full((nv,), 1 << 33, dtype=torch.int64, device=dev)
                   for _ in range(n_hash)]
            mxs = [torch.full((nv,), -1, dtype=torch.int64, device=dev)
                   for _ in range(n_hash)]
            for c0 in range(0, ne, CH):
                c1 = min(c0 + CH, ne)
                eidx = torch.arange(c0, c1, device=dev)
                seg_c = torch.searchsorted(rowptr, eidx, right=True) - 1
                del eidx
                tails_c = halo.tails_dense[c0:c1].to(torch.int64)
                tail_gid_c = gid_all[tails_c]
                cand = (tail_gid_c != (seg_c + base)) \
                    & uncolored_all[tails_c] & (colors[seg_c] < 0)
                del tails_c
                e_seg = seg_c[cand]
                e_tail_gid = tail_gid_c[cand]
                del seg_c, tail_gid_c, cand
                if not e_seg.numel():
                    continue
                for t in range(n_hash):
                    jh = _hash(e_tail_gid, seed + 1043 * t)
                    mns[t].scatter_reduce_(0, e_seg, jh, reduce="amin")
                    mxs[t].scatter_reduce_(0, e_seg, jh, reduce="amax")

        avail = torch.zeros(nv, 2 * n_hash, dtype=torch.bool, device=dev)
        vgid = torch.arange(base, dg.bound, device=dev)
        for t in range(n_hash):
            vh = _hash(vgid, seed + 1043 * t)
            avail[:, 2 * t] = vh < mns[t]
            avail[:, 2 * t + 1] = vh > mxs[t]

        uncolored = colors < 0
        navail = avail.sum(dim=1)
        can = uncolored & (navail > 0)
        if bool(can.any()):
            col_id = (vgid[can] % navail[can]).to(torch.int64)
            cum = torch.cumsum(avail[can].to(torch.int64), dim=1)
            slot = torch.argmax((cum == (col_id + 1).unsqueeze(1)).to(torch.int8),
                                dim=1)
            colors[can] = slot + next_color

        n_unassigned = int((colors < 0).sum())
        n_unassigned = int(comm.allreduce_scalar(float(n_unassigned)))
        current = tnv - n_unassigned
        next_color += 2 * n_hash
        seed = _hash_scalar(seed, 0)
        if current >= target or current == last_count:
            break
        last_count = current

    num_colors = next_color + 1
    colors = torch.where(colors < 0,
                         torch.tensor(num_colors - 1, device=dev), colors)
    return colors, num_colors


def check_coloring(dg: DistGraph, comm: Comm, colors: torch.Tensor,
                   exclude_color: int = -1, halo=None) -> int:
    """Count same-color adjacent pairs (ref distCheckColoring,
    coloring.cpp:447-593). `exclude_color`: skip pairs in this class (the
    uncolored-leftover overflow class is legitimately conflicting). Returns
    the global conflict count."""
    dev = dg.g.device
    nv = dg.nv
    if halo is None:
        halo = build_halo(dg, comm)
    ghost_colors = exchange_ghost_labels(halo, colors)
    call = torch.cat([colors, ghost_colors])
    gid_all = torch.cat([torch.arange(dg.base, dg.bound, device=dev), halo.ghosts])
    rowptr = dg.g.rowptr
    ne = dg.g.ne
    CH = 1 << 28  # chunked like distance1_coloring (>INT_MAX edge lists)
    conflicts = 0
    for c0 in range(0, ne, CH):
        c1 = min(c0 + CH, ne)
        eidx = torch.arange(c0, c1, device=dev)
        seg_c = torch.searchsorted(rowptr, eidx, right=True) - 1
        del eidx
        tails_c = halo.tails_dense[c0:c1].to(torch.int64)
        same = (gid_all[tails_c] != (seg_c + dg.base)) \
            & (call[tails_c] == colors[seg_c])
        if exclude_color >= 0:
            same &= colors[seg_c] != exclude_color
        conflicts += int(same.sum())
    return int(comm.allreduce_scalar(float(conflicts)))
