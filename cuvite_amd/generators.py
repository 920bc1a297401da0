"""In-memory graph generators: R-MAT (benchmark headline), RGG (reference
parity), the reproducible parallel LCG, and the Zachary karate-club test graph.

Reference analogs: generateInMemGraph/generateRGG (distgraph.cpp:341-933) and
the LCG parallel-prefix RNG (utils.hpp:76-271). Differences (deliberate,
MI355X-first):
  - The LCG stream is jumped ahead with modular exponentiation instead of the
    MPI 2x2 matrix parallel prefix: rank-independent, communication-free, and
    the SAME stream for any process count.
  - RGG vertex coordinates are pure functions of the global vertex id, so the
    generated graph is identical for any P (the reference's strip-local
    streams make the graph depend on P).
  - R-MAT (Graph500-style) is the headline generator (BASELINE.json configs);
    edges are generated in fixed global chunks seeded with counter-based
    Philox, so the edge list is identical for any P.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from .graph import Graph, DistGraph, Partition

# ---------------------------------------------------------------- LCG ------

MLCG = 2147483647  # 2^31 - 1
ALCG = 16807       # 7^5 (Park-Miller MINSTD), ref utils.hpp:25-27


def _seed_x0(seed: int) -> int:
    """Derive x0 from a 32-bit seed (ref reseeder uses std::seed_seq; we use
    splitmix32 — documented deviation, stream is still P-independent)."""
    z = (seed + 0x9E3779B9) & 0xFFFFFFFF
    z = ((z ^ (z >> 16)) * 0x85EBCA6B) & 0xFFFFFFFF
    z = ((z ^ (z >> 13)) * 0xC2B2AE35) & 0xFFFFFFFF
    z = z ^ (z >> 16)
    x0 = z % MLCG
    return x0 if x0 != 0 else 1


class LCG:
    """Park-Miller LCG with O(log k) jump-ahead: element k of the global
    stream is x0 * ALCG^k mod MLCG. Any rank can materialize any slice of the
    one global sequence without communication."""

    def __init__(self, seed: int = 1):
        self.x0 = _seed_x0(seed)

    def slice(self, start: int, count: int) -> np.ndarray:
        """Stream elements [start, start+count) as int64 in [0, MLCG)."""
        if count <= 0:
            return np.zeros(0, dtype=np.int64)
        a_start = pow(ALCG, start, MLCG)
        first = (self.x0 * a_start) % MLCG
        # powers table A^0..A^(count-1) by doubling (log2 numpy passes)
        pows = np.ones(1, dtype=np.int64)
        while pows.size < count:
            a_len = pow(ALCG, int(pows.size), MLCG)
            pows = np.concatenate([pows, (pows * a_len) % MLCG])
        pows = pows[:count]
        return (first * pows) % MLCG  # both < 2^31 so product < 2^62: no overflow

    def uniform(self, start: int, count: int) -> np.ndarray:
        """Stream elements mapped to [0,1) doubles (ref generate(), mult =
        1/MLCG)."""
        return self.slice(start, count).astype(np.float64) * (1.0 / MLCG)


# --------------------------------------------------------------- R-MAT -----

RMAT_A, RMAT_B, RMAT_C = 0.57, 0.19, 0.19  # Graph500 defaults (d = 0.05)
_CHUNK = 1 << 22  # global generation chunk (4M edges): P-independent unit


def _rmat_chunk(scale: int, n_edges: int, seed: int, chunk_id: int) -> tuple:
    """Generate `n_edges` undirected R-MAT edge tuples (u, v, w) for global
    chunk `chunk_id`, deterministically from (seed, chunk_id)."""
    rng = np.random.Generator(
        np.random.Philox(key=[(seed << 32) | 0xC0FFEE, chunk_id]))
    u = np.zeros(n_edges, dtype=np.int64)
    v = np.zeros(n_edges, dtype=np.int64)
    ab = RMAT_A + RMAT_B
    a_norm = RMAT_A / ab
    c_norm = RMAT_C / (1.0 - ab)
    for _ in range(scale):
        r1 = rng.random(n_edges)
        r2 = rng.random(n_edges)
        ubit = r1 > ab
        vbit = np.where(ubit, r2 > c_norm, r2 > a_norm)
        u = (u << 1) | ubit
        v = (v << 1) | vbit
    w = rng.random(n_edges)
    return u, v, w


def rmat_edges(scale: int, edgefactor: int, seed: int,
               part_lo: float = 0.0, part_hi: float = 1.0):
    """Undirected R-MAT edge tuples for the fraction [part_lo, part_hi) of the
    global edge list, chunk-deterministic: the union over disjoint fractions
    covering [0,1) equals the P=1 edge list exactly."""
    ne = edgefactor << scale
    nchunks = (ne + _CHUNK - 1) // _CHUNK
    c_lo = int(math.floor(part_lo * nchunks))
    c_hi = int(math.floor(part_hi * nchunks)) if part_hi < 1.0 else nchunks
    us, vs, ws = [], [], []
    for c in range(c_lo, c_hi):
        n_e = min(_CHUNK, ne - c * _CHUNK)
        u, v, w = _rmat_chunk(scale, n_e, seed, c)
        us.append(u)
        vs.append(v)
        ws.append(w)
    if not us:
        z = np.zeros(0, dtype=np.int64)
        return z, z, np.zeros(0, dtype=np.float64)
    return np.concatenate(us), np.concatenate(vs), np.concatenate(ws)


def rmat_edges_torch(scale: int, edgefactor: int, seed: int,
                     part_lo: float, part_hi: float, device,
                     weight_dtype: torch.dtype = torch.float64):
    """GPU-resident R-MAT edge generation, chunk-deterministic like
    rmat_edges() but using torch's Philox on `device` (one seeded generator
    per global chunk, so the graph is identical for any process count on the
    same device type). Returns (u, v, w) undirected tuples on `device`."""
    ne = edgefactor << scale
    nchunks = (ne + _CHUNK - 1) // _CHUNK
    c_lo = int(math.floor(part_lo * nchunks))
    c_hi = int(math.floor(part_hi * nchunks)) if part_hi < 1.0 else nchunks
    ab = RMAT_A + RMAT_B
    a_norm = RMAT_A / ab
    c_norm = RMAT_C / (1.0 - ab)
    us, vs, ws = [], [], []
    gen = torch.Generator(device=device)
    for c in range(c_lo, c_hi):
        n_e = min(_CHUNK, ne - c * _CHUNK)
        gen.manual_seed((seed * 0x1F123BB5 + c) & 0x7FFFFFFFFFFF)
        u = torch.zeros(n_e, dtype=torch.int64, device=device)
        v = torch.zeros(n_e, dtype=torch.int64, device=device)
        for _ in range(scale):
            r1 = torch.rand(n_e, generator=gen, device=device)
            r2 = torch.rand(n_e, generator=gen, device=device)
            ubit = r1 > ab
            vbit = torch.where(ubit, r2 > c_norm, r2 > a_norm)
            u = (u << 1) | ubit
            v = (v << 1) | vbit
        w = torch.rand(n_e, generator=gen, device=device, dtype=weight_dtype)
        us.append(u)
        vs.append(v)
        ws.append(w)
    if not us:
        z = torch.zeros(0, dtype=torch.int64, device=device)
        return z, z, torch.zeros(0, dtype=weight_dtype, device=device)
    return torch.cat(us), torch.cat(vs), torch.cat(ws)


_ROUTE_CHUNK = 1 << 28  # 256M tuples per routing slice (torch op INT_MAX cap)


def rmat_dist_graph(scale: int, edgefactor: int, seed: int, comm,
                    device, weight_dtype: torch.dtype = torch.float64):
    """This rank's shard of the symmetrized R-MAT graph: generate the rank's
    slice of the global undirected edge list, emit both directed copies, and
    route each to its 1-D owner. Routing is chunked owner-bucketing (no
    global argsort: torch.sort is capped at INT_MAX elements, and a scale-26
    symmetrized list is 2.1e9 tuples)."""
    from .graph import DistGraph, Graph, Partition

    nv = 1 << scale
    part = Partition.contiguous(nv, comm.world)
    lo = comm.rank / comm.world
    hi = (comm.rank + 1) / comm.world
    if device.type == "cuda":
        u, v, w = rmat_edges_torch(scale, edgefactor, seed, lo, hi, device,
                                   weight_dtype=weight_dtype)
    else:
        uu, vv, ww = rmat_edges(scale, edgefactor, seed, lo, hi)
        u, v = torch.from_numpy(uu), torch.from_numpy(vv)
        w = torch.from_numpy(ww).to(weight_dtype)

    import os as _os
    import sys as _sys
    _dbg = _os.environ.get("CUVITE_PROGRESS")

    def _p(m):
        if _dbg:
            print(f"[rmat] {m}", file=_sys.stderr, flush=True)

    if comm.world == 1:
        _p(f"edges generated: {u.numel()}")
        src = torch.cat([u, v])
        dst = torch.cat([v, u])
        ww = torch.cat([w, w])
        del u, v, w
        _p("symmetrized; building CSR")
        g = Graph.from_edge_tuples(nv, src, dst, ww)
        _p("CSR done")
        return DistGraph(g, part, comm.rank)

    parts_dev = part.parts.to(device)
    kept_s, kept_d, kept_w = [], [], []
    n = u.numel()
    # every rank must run the same number of collective rounds
    n_max = int(comm.allreduce_scalar(float(n), op="max"))
    for c0 in range(0, max(n_max, 1), _ROUTE_CHUNK):
        c1 = min(c0 + _ROUTE_CHUNK, n)
        c0 = min(c0, c1)
        # both directed copies of this chunk
        cs = torch.cat([u[c0:c1], v[c0:c1]])
        cd = torch.cat([v[c0:c1], u[c0:c1]])
        cw = torch.cat([w[c0:c1], w[c0:c1]])
        owner = torch.searchsorted(parts_dev[1:], cs, right=True)
        send_s, send_d, send_w = [], [], []
        for p in range(comm.world):
            m = owner == p
            send_s.append(cs[m])
            send_d.append(cd[m])
            send_w.append(cw[m])
        got_s = comm.all_to_all_v(send_s)
        cnts = [int(t.numel()) for t in got_s]
        got_d = comm.all_to_all_v(send_d, cnts)
        got_w = comm.all_to_all_v(send_w, cnts)
        kept_s.append(torch.cat(got_s))
        kept_d.append(torch.cat(got_d))
        kept_w.append(torch.cat(got_w))
    del u, v, w
    src = torch.cat(kept_s) if kept_s else torch.zeros(0, dtype=torch.int64,
                                                       device=device)
    del kept_s
    dst = torch.cat(kept_d) if kept_d else torch.zeros(0, dtype=torch.int64,
                                                       device=device)
    del kept_d
    ww = torch.cat(kept_w) if kept_w else torch.zeros(0, dtype=weight_dtype,
                                                      device=device)
    del kept_w
    g = Graph.from_edge_tuples(part.nv_local(comm.rank), src, dst, ww,
                               base=part.base(comm.rank))
    return DistGraph(g, part, comm.rank)


def rmat_graph(scale: int, edgefactor: int = 16, seed: int = 1,
               weight_dtype: torch.dtype = torch.float64) -> Graph:
    """Whole symmetrized R-MAT graph on one process (tests / small runs)."""
    u, v, w = rmat_edges(scale, edgefactor, seed)
    nv = 1 << scale
    src = torch.from_numpy(np.concatenate([u, v]))
    dst = torch.from_numpy(np.concatenate([v, u]))
    ww = torch.from_numpy(np.concatenate([w, w])).to(weight_dtype)
    return Graph.from_edge_tuples(nv, src, dst, ww)


# ----------------------------------------------------------------- LFR -----

def lfr_graph(nv: int, mu: float = 0.4, avg_deg: int = 20,
              max_deg: Optional[int] = None, tau1: float = 2.5,
              tau2: float = 1.5, min_comm: Optional[int] = None,
              max_comm: Optional[int] = None, seed: int = 1,
              weight_dtype: torch.dtype = torch.float64):
    """LFR-style benchmark graph with planted communities.

    The reference consumes LFR graphs as external files with a ground-truth
    community file (-g, README:105-117; loadGroundTruthFile louvain.cpp:
    3272-3304); this generator synthesizes an equivalent benchmark in-process
    (no network access for the canonical LFR binaries): power-law degrees
    (exponent tau1), power-law community sizes (exponent tau2), and mixing
    parameter mu = fraction of each vertex's edges that leave its community.

    Returns (Graph symmetrized, ground_truth int64 [nv]).
    """
    rng = np.random.Generator(np.random.Philox(key=[(seed << 32) | 0x1F12,
                                                    nv & 0x7FFFFFFF]))
    if max_deg is None:
        max_deg = max(avg_deg * 5, int(math.sqrt(nv)))
    min_deg = max(2, int(avg_deg * (tau1 - 2) / (tau1 - 1)))  # mean target
    if min_comm is None:
        min_comm = max(10, max_deg // 2)
    if max_comm is None:
        max_comm = max(min_comm + 1, nv // 10)

    def powlaw(n, lo, hi, tau):
        """n samples from P(x) ~ x^-tau on [lo, hi] (inverse transform)."""
        u = rng.random(n)
        a = 1.0 - tau
        lo_a, hi_a = lo ** a, hi ** a
        return np.floor((lo_a + u * (hi_a - lo_a)) ** (1.0 / a)).astype(
            np.int64)

    # community sizes until they cover nv
    sizes = []
    tot = 0
    while tot < nv:
        s = int(powlaw(1, min_comm, max_comm, tau2)[0])
        sizes.append(s)
        tot += s
    sizes[-1] -= tot - nv  # trim the last one
    if sizes[-1] < 2 and len(sizes) > 1:
        sizes[-2] += sizes[-1]
        sizes.pop()
    sizes = np.array(sizes, dtype=np.int64)
    ncomm = len(sizes)
    truth = np.repeat(np.arange(ncomm, dtype=np.int64), sizes)
    perm = rng.permutation(nv)
    truth = truth[np.argsort(perm)]  # random vertex->community assignment

    deg = powlaw(nv, min_deg, max_deg, tau1)
    # cap intra-degree by community size - 1
    comm_size_of = sizes[truth]
    k_in = np.minimum(np.rint(deg * (1.0 - mu)).astype(np.int64),
                      comm_size_of - 1)
    k_out = deg - k_in

    srcs, dsts = [], []

    # intra-community configuration model per community
    order = np.argsort(truth, kind="stable")
    bounds = np.searchsorted(truth[order], np.arange(ncomm + 1))
    for c in range(ncomm):
        members = order[bounds[c]:bounds[c + 1]]
        stubs = np.repeat(members, k_in[members])
        if stubs.size < 2:
            continue
        rng.shuffle(stubs)
        if stubs.size % 2:
            stubs = stubs[:-1]
        half = stubs.size // 2
        a, b = stubs[:half], stubs[half:]
        m = a != b
        srcs.append(a[m])
        dsts.append(b[m])

    # inter-community stub pairing (reject pairs landing intra)
    stubs = np.repeat(np.arange(nv, dtype=np.int64), k_out)
    rng.shuffle(stubs)
    if stubs.size % 2:
        stubs = stubs[:-1]
    half = stubs.size // 2
    a, b = stubs[:half], stubs[half:]
    m = (a != b) & (truth[a] != truth[b])
    srcs.append(a[m])
    dsts.append(b[m])

    u = np.concatenate(srcs)
    v = np.concatenate(dsts)
    # dedup undirected pairs
    lo_ = np.minimum(u, v)
    hi_ = np.maximum(u, v)
    key = lo_ * nv + hi_
    key = np.unique(key)
    u = key // nv
    v = key % nv
    w = rng.uniform(0.01, 1.0, u.size)

    src = torch.from_numpy(np.concatenate([u, v]))
    dst = torch.from_numpy(np.concatenate([v, u]))
    ww = torch.from_numpy(np.concatenate([w, w])).to(weight_dtype)
    g = Graph.from_edge_tuples(nv, src, dst, ww)
    return g, torch.from_numpy(truth)


def lfr_dist_graph(nv: int, rank: int, nranks: int, mu: float = 0.4,
                   seed: int = 1, device=None, **kw):
    """This rank's shard of an LFR benchmark (generated identically on every
    rank from the seed, then sliced by the contiguous partition). LFR
    acceptance configs are ~1M vertices — whole-graph generation per rank is
    cheap next to 288 GB HBM. Returns (DistGraph, ground_truth_global)."""
    from .graph import DistGraph, Graph as _G, Partition
    g, truth = lfr_graph(nv, mu=mu, seed=seed, **kw)
    part = Partition.contiguous(nv, nranks)
    base, bound = part.base(rank), part.bound(rank)
    e0, e1 = int(g.rowptr[base]), int(g.rowptr[bound])
    lg = _G(g.rowptr[base:bound + 1] - g.rowptr[base],
            g.tails[e0:e1], g.weights[e0:e1])
    dg = DistGraph(lg, part, rank)
    if device is not None:
        dg = dg.to(device)
    return dg, truth


# ----------------------------------------------------------------- RGG -----

def rgg_radius(nv: int) -> float:
    """Reference cutoff: mean of the connectivity threshold sqrt(ln nv/(pi nv))
    and sqrt(2.0736/nv) (distgraph.cpp:344-349)."""
    rc = math.sqrt(math.log(nv) / (math.pi * nv))
    rt = math.sqrt(2.0736 / nv)
    return 0.5 * (rc + rt)


def _rgg_coords(lcg: LCG, lo: int, hi: int, nv: int):
    """Coordinates of global vertices [lo, hi): x = u_{2i} in [0,1),
    y = (i + u_{2i+1})/nv — ids are ordered along y, so contiguous id ranges
    are y-bands and any rank can regenerate any window without communication."""
    u = lcg.uniform(2 * lo, 2 * (hi - lo))
    x = u[0::2]
    y = (np.arange(lo, hi, dtype=np.float64) + u[1::2]) / nv
    return x, y


def rgg_local_edges(nv: int, rank: int, nranks: int, seed: int = 1,
                    random_edge_percent: float = 0.0):
    """Directed edge tuples (src global, dst global, weight) for all src owned
    by `rank` under the contiguous partition. Euclidean-distance weights
    (distgraph.cpp:361-362); optional extra random edges with U(0.01, 1.0)
    weights (distgraph.cpp:757)."""
    part = Partition.contiguous(nv, nranks)
    base, bound = part.base(rank), part.bound(rank)
    rn = rgg_radius(nv)
    lcg = LCG(seed)
    halo = int(math.ceil(rn * nv)) + 1
    lo, hi = max(0, base - halo), min(nv, bound + halo)
    x, y = _rgg_coords(lcg, lo, hi, nv)
    ids = np.arange(lo, hi, dtype=np.int64)

    # spatial hash on cells of size rn; candidate pairs from 3x3 neighborhoods
    ncell = max(1, int(1.0 / rn))
    cx = np.minimum((x * ncell).astype(np.int64), ncell - 1)
    cy = np.minimum((y * ncell).astype(np.int64), ncell - 1)
    cell = cy * ncell + cx
    order = np.argsort(cell, kind="stable")
    cell_s = cell[order]
    starts = np.searchsorted(cell_s, np.arange(ncell * ncell))
    ends = np.searchsorted(cell_s, np.arange(ncell * ncell), side="right")

    is_mine = (ids >= base) & (ids < bound)
    srcs, dsts, wts = [], [], []
    mine_idx = np.nonzero(is_mine)[0]
    # gather candidates per 3x3 neighborhood of each cell containing my points
    my_cells = np.unique(cell[mine_idx])
    for c in my_cells:
        cyy, cxx = divmod(int(c), ncell)
        cand = []
        for dy in (-1, 0, 1):
            for dx in (-1, 0, 1):
                yy, xx = cyy + dy, cxx + dx
                if 0 <= yy < ncell and 0 <= xx < ncell:
                    c2 = yy * ncell + xx
                    cand.append(order[starts[c2]:ends[c2]])
        cand = np.concatenate(cand) if cand else np.zeros(0, dtype=np.int64)
        pts = order[starts[int(c)]:ends[int(c)]]
        pts = pts[is_mine[pts]]
        if pts.size == 0 or cand.size == 0:
            continue
        dxm = x[pts][:, None] - x[cand][None, :]
        dym = y[pts][:, None] - y[cand][None, :]
        d = np.sqrt(dxm * dxm + dym * dym)
        pi, ci = np.nonzero((d <= rn) & (ids[pts][:, None] != ids[cand][None, :]))
        srcs.append(ids[pts][pi])
        dsts.append(ids[cand][ci])
        wts.append(d[pi, ci])

    if srcs:
        src = np.concatenate(srcs)
        dst = np.concatenate(dsts)
        w = np.concatenate(wts)
    else:
        src = np.zeros(0, dtype=np.int64)
        dst = np.zeros(0, dtype=np.int64)
        w = np.zeros(0, dtype=np.float64)

    if random_edge_percent > 0.0:
        # reference: ceil(percent * ne_global / 100) extra undirected random
        # edges (distgraph.cpp:706-760). P-independent: both endpoints drawn
        # from a dedicated Philox stream; each rank emits the directed copies
        # whose src it owns.
        ne_base_local = src.size
        # estimate global count deterministically: use expected edges via a
        # fixed chunked draw keyed only by (seed, nv)
        rng = np.random.Generator(
            np.random.Philox(key=[(seed << 32) | 0xFACADE, nv & 0x7FFFFFFF]))
        n_rand = int(math.ceil(random_edge_percent * nv / 100.0))
        ru = rng.integers(0, nv, n_rand)
        rv = rng.integers(0, nv, n_rand)
        rw = rng.uniform(0.01, 1.0, n_rand)
        for a, b in ((ru, rv), (rv, ru)):
            m = (a >= base) & (a < bound) & (a != b)
            src = np.concatenate([src, a[m]])
            dst = np.concatenate([dst, b[m]])
            w = np.concatenate([w, rw[m]])

    return src, dst, w


def rgg_dist_graph(nv: int, rank: int, nranks: int, seed: int = 1,
                   random_edge_percent: float = 0.0,
                   weight_dtype: torch.dtype = torch.float64) -> DistGraph:
    src, dst, w = rgg_local_edges(nv, rank, nranks, seed, random_edge_percent)
    part = Partition.contiguous(nv, nranks)
    g = Graph.from_edge_tuples(part.nv_local(rank), torch.from_numpy(src),
                               torch.from_numpy(dst),
                               torch.from_numpy(w).to(weight_dtype),
                               base=part.base(rank))
    return DistGraph(g, part, rank)


# ------------------------------------------------------------- karate ------

_KARATE_EDGES = [
    (0, 1), (0, 2), (0, 3), (0, 4), (0, 5), (0, 6), (0, 7), (0, 8), (0, 10),
    (0, 11), (0, 12), (0, 13), (0, 17), (0, 19), (0, 21), (0, 31), (1, 2),
    (1, 3), (1, 7), (1, 13), (1, 17), (1, 19), (1, 21), (1, 30), (2, 3),
    (2, 7), (2, 8), (2, 9), (2, 13), (2, 27), (2, 28), (2, 32), (3, 7),
    (3, 12), (3, 13), (4, 6), (4, 10), (5, 6), (5, 10), (5, 16), (6, 16),
    (8, 30), (8, 32), (8, 33), (9, 33), (13, 33), (14, 32), (14, 33),
    (15, 32), (15, 33), (18, 32), (18, 33), (19, 33), (20, 32), (20, 33),
    (22, 32), (22, 33), (23, 25), (23, 27), (23, 29), (23, 32), (23, 33),
    (24, 25), (24, 27), (24, 31), (25, 31), (26, 29), (26, 33), (27, 33),
    (28, 31), (28, 33), (29, 32), (29, 33), (30, 32), (30, 33), (31, 32),
    (31, 33), (32, 33),
]


def karate_graph(weight_dtype: torch.dtype = torch.float64) -> Graph:
    """Zachary karate club (34 vertices, 78 undirected edges, unit weights) —
    the reference's canonical smoke-test graph (README:60)."""
    e = torch.tensor(_KARATE_EDGES, dtype=torch.int64)
    src = torch.cat([e[:, 0], e[:, 1]])
    dst = torch.cat([e[:, 1], e[:, 0]])
    w = torch.ones(src.numel(), dtype=weight_dtype)
    return Graph.from_edge_tuples(34, src, dst, w)


def _reorder_rows(g, order, inv_or_map, chunk: int = 1 << 28):
    """Physically permute a Graph's rows to `order` (new row p = old row
    order[p]) while mapping each tail id through `inv_or_map` (old global
    id -> new global id, indexable by the old tail values). Chunked: gathers
    over >INT_MAX edges overflow torch indexing."""
    from .graph import Graph
    dev = g.device
    nv = g.nv
    deg = g.degrees()
    new_deg = deg[order]
    rowptr2 = torch.zeros(nv + 1, dtype=torch.int64, device=dev)
    rowptr2[1:] = torch.cumsum(new_deg, dim=0)
    ne = g.ne
    tails2 = torch.empty(ne, dtype=torch.int64, device=dev)
    weights2 = torch.empty(ne, dtype=g.weights.dtype, device=dev)
    old_start = g.rowptr[order]  # old CSR start per NEW row
    row_lo = 0
    while row_lo < nv:
        row_hi = int(torch.searchsorted(
            rowptr2, torch.tensor(int(rowptr2[row_lo]) + chunk,
                                  device=dev)))
        row_hi = max(row_lo + 1, min(row_hi, nv))
        e0 = int(rowptr2[row_lo])
        e1 = int(rowptr2[row_hi])
        n = e1 - e0
        if n:
            seg = torch.searchsorted(
                rowptr2, torch.arange(e0, e1, device=dev), right=True) - 1
            eidx = old_start[seg] + \
                (torch.arange(e0, e1, device=dev) - rowptr2[seg])
            del seg
            tails2[e0:e1] = inv_or_map[g.tails[eidx]]
            weights2[e0:e1] = g.weights[eidx]
            del eidx
        row_lo = row_hi
    return Graph(rowptr2, tails2, weights2)


def degree_sort_dist(dg, comm, return_maps: bool = False):
    """Per-rank-range degree-descending relabeling of a DistGraph: each rank
    permutes vertices WITHIN its own contiguous [base, bound) range (owners,
    partition and edge balance unchanged), so new gid of old vertex v =
    base(owner) + rank-of-v-by-degree-within-owner. Tails are mapped through
    a one-time allgathered global old->new table (nv_global * 8 B; 0.5 GB at
    s26 — cheap in 288 GB HBM). Same locality rationale as
    degree_sort_graph: each rank's hubs pack at the front of its range, so
    the per-edge curr_comm gathers concentrate into per-rank hot prefixes.
    NOTE: the permutation is P-dependent (each P yields a different but
    isomorphic labeling), so cross-P trajectories are not comparable
    bit-for-bit — per-iteration TEPS and converged Q are unaffected."""
    from .graph import DistGraph
    dev = dg.g.device
    nv = dg.nv
    base = dg.base
    deg = dg.g.degrees()
    order = torch.argsort(-deg, stable=True)
    inv = torch.empty(nv, dtype=torch.int64, device=dev)
    inv[order] = torch.arange(nv, device=dev)
    local_map = base + inv  # old local v -> new gid
    if comm.world == 1:
        gmap = local_map
    else:
        import torch.distributed as dist
        nmax = max(dg.partition.nv_local(p) for p in range(comm.world))
        pad = torch.full((nmax,), -1, dtype=torch.int64, device=dev)
        pad[:nv] = local_map
        outs = [torch.empty(nmax, dtype=torch.int64, device=dev)
                for _ in range(comm.world)]
        dist.all_gather(outs, pad)
        gmap = torch.cat([outs[p][:dg.partition.nv_local(p)]
                          for p in range(comm.world)])
    g2 = _reorder_rows(dg.g, order, gmap)
    dg2 = DistGraph(g2, dg.partition, dg.rank)
    if return_maps:
        # order: new local position -> old local position;
        # gmap: old global id -> new global id
        return dg2, order, gmap
    return dg2


def degree_sort_graph(g, chunk: int = 1 << 28):
    """Isomorphic relabeling of a single-rank Graph with vertices ordered by
    degree (descending): vertex new-id p = old vertex order[p]. Returns
    (new Graph, order) where order maps new ids back to old ids.

    Locality optimization for power-law graphs: the local-move kernels'
    dominant traffic is the data-dependent curr_comm[tail] gather (PMC: ~80%
    SQ_WAIT_ANY, profiles/round2_kernel_stats.md); hubs receive a large
    fraction of all edges, so packing them at the front of the id space
    turns those gathers into hits on a few hundred KB that stay resident in
    the per-XCD L2s. Measured at R-MAT s26: 120.0 -> 97.8 ms/step (+23%,
    gpurun r2d). The graph is unchanged up to isomorphism — same degrees,
    weights, modularity landscape; community labels map through `order`."""
    dev = g.device
    nv = g.nv
    order = torch.argsort(-g.degrees(), stable=True)
    inv = torch.empty(nv, dtype=torch.int64, device=dev)
    inv[order] = torch.arange(nv, device=dev)
    return _reorder_rows(g, order, inv, chunk), order
