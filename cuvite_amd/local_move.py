"""One Louvain local-moving iteration, in dense index space.

This module is the semantic specification of the flagship HIP kernel and the
CPU oracle. Semantics transcribed from the reference CPU path
(distBuildLocalMapCounter louvain.cpp:2384-2431, distGetMaxIndex
louvain.cpp:2185-2246, distExecuteLouvainIteration louvain.cpp:2246-2384):

  For each local vertex i (dense id in [0, nv)):
    - build the map {community -> sum of edge weights from i}, over all edges;
      a self-edge contributes to both selfLoop and the map entry of i's own
      community cc;
    - clusterWeight[i] = map[cc] (0 if absent);
    - eix = map[cc] - selfLoop, ax = degree(cc) - vDegree[i];
    - for every candidate y != cc: gain = 2*(map[y] - eix)
        - 2*vDegree[i]*(degree(y) - ax)*constant;
      pick max gain (> 0 strictly), ties -> smallest GLOBAL community id;
    - singleton-swap guard: if size(chosen) == 1 and size(cc) == 1 and
      gid(chosen) > gid(cc), stay (louvain.cpp:2238-2239);
    - isolated vertices (no edges) stay.

Dense index space: local vertices own dense ids [0, nv); ghost vertices get
[nv, nv+ng); local communities own dense ids [0, nv) (community c owned by
this rank maps to c - base); referenced remote communities get [nv, nv+nrc).
`comm_gid[dense]` maps a dense community id back to its global id (used only
for tie-breaks and the singleton guard).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class MoveInputs:
    """Inputs for one local-move iteration (all tensors on one device)."""

    rowptr: torch.Tensor       # int64 [nv+1]
    tails: torch.Tensor        # int32/int64 [ne], dense VERTEX ids in [0, nv+ng)
    weights: torch.Tensor      # W [ne]
    curr_comm: torch.Tensor    # int32/int64 [nv+ng], dense COMMUNITY ids in [0, nc)
    v_degree: torch.Tensor     # W [nv]
    comm_size: torch.Tensor    # int64 [nc]
    comm_degree: torch.Tensor  # W [nc]
    comm_gid: torch.Tensor     # int64 [nc]  dense comm id -> global comm id
    constant: float            # 1 / (2m)

    @property
    def nv(self) -> int:
        return self.rowptr.numel() - 1


def local_move_pydict(inp: MoveInputs):
    """Direct dict-based transcription of the reference semantics. Slow;
    tests-only golden model."""
    nv = inp.nv
    rowptr = inp.rowptr.tolist()
    tails = inp.tails.tolist()
    weights = inp.weights.tolist()
    curr = inp.curr_comm.tolist()
    vdeg = inp.v_degree.tolist()
    csize = inp.comm_size.tolist()
    cdeg = inp.comm_degree.tolist()
    gid = inp.comm_gid.tolist()
    const = inp.constant

    target = [0] * nv
    cluster_weight = [0.0] * nv
    for i in range(nv):
        cc = curr[i]
        e0, e1 = rowptr[i], rowptr[i + 1]
        if e0 == e1:
            target[i] = cc
            continue
        selfloop = 0.0
        counter = {cc: 0.0}
        for k in range(e0, e1):
            t = tails[k]
            w = weights[k]
            if t == i:
                selfloop += w
            tc = curr[t]
            counter[tc] = counter.get(tc, 0.0) + w
        cluster_weight[i] = counter[cc]
        eix = counter[cc] - selfloop
        ax = cdeg[cc] - vdeg[i]
        max_gain = 0.0
        max_idx = cc
        max_size = csize[cc]
        for y, eiy in counter.items():
            if y == cc:
                continue
            gain = 2.0 * (eiy - eix) - 2.0 * vdeg[i] * (cdeg[y] - ax) * const
            if gain > max_gain or (gain == max_gain and gain != 0.0
                                   and gid[y] < gid[max_idx]):
                max_gain = gain
                max_idx = y
                max_size = csize[y]
        if max_size == 1 and csize[cc] == 1 and gid[max_idx] > gid[cc]:
            max_idx = cc
        target[i] = max_idx

    dev = inp.rowptr.device
    return (torch.tensor(target, dtype=inp.curr_comm.dtype, device=dev),
            torch.tensor(cluster_weight, dtype=inp.weights.dtype, device=dev))


def local_move_torch(inp: MoveInputs):
    """Vectorized torch implementation of the same op (CPU fast path and the
    plain-PyTorch reference the HIP kernel is tested against).

    Returns (target [nv] dense comm ids, cluster_weight [nv])."""
    nv = inp.nv
    dev = inp.rowptr.device
    W = inp.weights.dtype
    deg = inp.rowptr[1:] - inp.rowptr[:-1]
    ne = inp.tails.numel()
    if ne == 0:
        return inp.curr_comm[:nv].clone(), torch.zeros(nv, dtype=W, device=dev)

    seg = torch.repeat_interleave(torch.arange(nv, device=dev), deg)  # src per edge
    tails = inp.tails.to(torch.int64)
    tcomm = inp.curr_comm.to(torch.int64)[tails]                      # comm per edge

    # self-loop weight per vertex
    selfloop = torch.zeros(nv, dtype=W, device=dev)
    self_mask = tails == seg
    if bool(self_mask.any()):
        selfloop.index_add_(0, seg[self_mask], inp.weights[self_mask])

    # per-(vertex, community) weight sums via sort + unique_consecutive
    nc = inp.comm_size.numel()
    key = seg * nc + tcomm
    key_s, order = torch.sort(key)
    w_s = inp.weights[order]
    uniq, inv = torch.unique_consecutive(key_s, return_inverse=True)
    gsum = torch.zeros(uniq.numel(), dtype=W, device=dev)
    gsum.index_add_(0, inv, w_s)
    gv = uniq // nc     # vertex of each group
    gc = uniq % nc      # community of each group

    cc = inp.curr_comm.to(torch.int64)[:nv]
    own = gc == cc[gv]
    cluster_weight = torch.zeros(nv, dtype=W, device=dev)
    cluster_weight[gv[own]] = gsum[own]

    eix = cluster_weight - selfloop
    ax = inp.comm_degree[cc] - inp.v_degree

    cand = ~own
    cv, cy, eiy = gv[cand], gc[cand], gsum[cand]
    # gain arithmetic always in fp64 (the HIP kernels do the same regardless
    # of the weight dtype, so fp32-weight tie-breaks stay aligned)
    gain = (2.0 * (eiy.to(torch.float64) - eix[cv].to(torch.float64))
            - 2.0 * inp.v_degree[cv].to(torch.float64)
            * (inp.comm_degree[cy].to(torch.float64)
               - ax[cv].to(torch.float64)) * inp.constant)

    target = cc.clone()
    if cv.numel():
        # per-vertex argmax with (max gain, then min GLOBAL id) tie-break:
        # stable sort by gid asc, then by gain desc, then by vertex asc
        gidy = inp.comm_gid[cy]
        o1 = torch.argsort(gidy, stable=True)
        o2 = o1[torch.argsort(-gain[o1].to(torch.float64), stable=True)]
        o3 = o2[torch.argsort(cv[o2], stable=True)]
        cv3 = cv[o3]
        first = torch.ones_like(cv3, dtype=torch.bool)
        first[1:] = cv3[1:] != cv3[:-1]
        sel = o3[first]
        v_sel, y_sel, g_sel = cv[sel], cy[sel], gain[sel]
        pos = g_sel > 0
        v_sel, y_sel = v_sel[pos], y_sel[pos]
        # singleton guard (on global ids)
        guard = ((inp.comm_size[y_sel] == 1) & (inp.comm_size[cc[v_sel]] == 1)
                 & (inp.comm_gid[y_sel] > inp.comm_gid[cc[v_sel]]))
        keep = ~guard
        target[v_sel[keep]] = y_sel[keep]

    return target.to(inp.curr_comm.dtype), cluster_weight


def modularity_parts(cluster_weight: torch.Tensor,
                     local_comm_degree: torch.Tensor) -> torch.Tensor:
    """Local (Sum e_xx, Sum a_x^2) pair in fp64 (ref distComputeModularity,
    louvain.cpp:2433-2482). Caller allreduces and applies
    Q = e*c - a2*c^2."""
    le = cluster_weight.to(torch.float64).sum()
    la2 = (local_comm_degree.to(torch.float64) ** 2).sum()
    return torch.stack([le, la2])
