"""Graph substrate: local CSR graph + 1-D distributed partition.

Semantics follow the reference cuVite graph layer (graph.hpp:27-57,
distgraph.hpp:27-237) but the representation is torch tensors so the same
objects move between CPU (tests / oracle) and MI355X HBM unchanged.

Conventions:
  - Vertex ids are global int64 in [0, nv_global).
  - A rank owns the contiguous range [base, bound) given by the partition.
  - The local CSR stores, for each local vertex, its full adjacency with
    *global* tail ids and weights; an undirected edge (u, v) appears as
    u->v and v->u (each stored on its endpoint's owner), matching the Vite
    binary format.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class Partition:
    """1-D contiguous vertex partition: rank p owns [parts[p], parts[p+1]).

    Mirrors the reference's `parts` prefix array (distgraph.hpp:211-222);
    ownership lookup is a binary search over the prefix boundaries.
    """

    parts: torch.Tensor  # int64, shape [nranks+1], parts[0]=0, parts[-1]=nv_global

    @staticmethod
    def contiguous(nv_global: int, nranks: int) -> "Partition":
        """Even vertex split: parts[p] = floor(nv*p/P) (distgraph.cpp:118-119)."""
        p = torch.arange(nranks + 1, dtype=torch.int64)
        return Partition((p * nv_global) // nranks)

    @staticmethod
    def from_bounds(bounds) -> "Partition":
        t = torch.as_tensor(bounds, dtype=torch.int64)
        assert t[0] == 0 and torch.all(t[1:] >= t[:-1])
        return Partition(t)

    @staticmethod
    def edge_balanced(index: torch.Tensor, nranks: int) -> "Partition":
        """Edge-balanced ranges from a global CSR offset array (ref: balanceEdges,
        distgraph.cpp:22-66): choose boundaries so each rank holds ~ne/P edges."""
        nv = index.numel() - 1
        ne = int(index[-1])
        targets = (torch.arange(1, nranks, dtype=torch.float64) * ne / nranks).to(torch.int64)
        cuts = torch.searchsorted(index, targets, right=False).clamp_(0, nv)
        parts = torch.cat(
            [torch.zeros(1, dtype=torch.int64), cuts.to(torch.int64),
             torch.tensor([nv], dtype=torch.int64)]
        )
        # enforce monotonicity (tiny graphs / skewed prefix)
        parts = torch.cummax(parts, dim=0).values
        return Partition(parts)

    @property
    def nranks(self) -> int:
        return self.parts.numel() - 1

    @property
    def nv_global(self) -> int:
        return int(self.parts[-1])

    def base(self, rank: int) -> int:
        return int(self.parts[rank])

    def bound(self, rank: int) -> int:
        return int(self.parts[rank + 1])

    def nv_local(self, rank: int) -> int:
        return self.bound(rank) - self.base(rank)

    def owner(self, vids: torch.Tensor) -> torch.Tensor:
        """Owner rank of each global vertex id (vectorized)."""
        parts = self.parts.to(vids.device)
        return torch.searchsorted(parts, vids, right=True) - 1

    def owner_one(self, v: int) -> int:
        return int(self.owner(torch.tensor([v], dtype=torch.int64))[0])


class Graph:
    """Local CSR graph: rowptr[nv+1] (int64), tails[ne] (global int64 ids),
    weights[ne] (fp32/fp64). Reference: Graph, graph.hpp:27-57."""

    def __init__(self, rowptr: torch.Tensor, tails: torch.Tensor, weights: torch.Tensor):
        assert rowptr.dtype == torch.int64 and tails.dtype == torch.int64
        assert rowptr.numel() >= 1 and int(rowptr[0]) == 0
        assert tails.numel() == weights.numel() == int(rowptr[-1])
        self.rowptr = rowptr
        self.tails = tails
        self.weights = weights
        self._ne_released: int = -1  # set by release_tails()

    @property
    def nv(self) -> int:
        return self.rowptr.numel() - 1

    @property
    def ne(self) -> int:
        if self._ne_released >= 0:
            return self._ne_released
        return self.tails.numel()

    def release_tails(self):
        """Free the global-id tails array while keeping `ne` valid. Once a
        HaloContext exists, every compute path reads its dense int32 tails
        instead — the int64 originals are 34 GB of dead weight at R-MAT s27
        and were the difference between the converged multi-phase run
        fitting in 288 GB HBM or not. After release, ghost discovery /
        fresh halo builds on this graph raise."""
        self._ne_released = self.tails.numel()
        self.tails = torch.empty(0, dtype=torch.int64, device=self.device)

    @property
    def device(self):
        return self.rowptr.device

    def degrees(self) -> torch.Tensor:
        return self.rowptr[1:] - self.rowptr[:-1]

    def to(self, device) -> "Graph":
        return Graph(self.rowptr.to(device), self.tails.to(device), self.weights.to(device))

    def edge_range(self, v: int):
        return int(self.rowptr[v]), int(self.rowptr[v + 1])

    @staticmethod
    def from_edge_tuples(nv: int, src: torch.Tensor, dst: torch.Tensor,
                         w: torch.Tensor, base: int = 0) -> "Graph":
        """Assemble a CSR from (src, dst, w) directed edge tuples where src are
        global ids in [base, base+nv). Reference analog: processGraphData
        (utils.cpp:10-87). On GPU this uses the sort-free HIP builder
        (histogram + atomic-cursor placement; torch.argsort is capped at
        INT_MAX elements and a lexicographic sort is not needed — all CSR
        consumers are row-order-invariant). The CPU path sorts by (src, dst)
        for determinism.

        Determinism note: the GPU builder leaves
        row-internal edge order nondeterministic, so fp reductions over rows
        (row_sum, the hub-merge cumsum) are reproducible only per CSR
        instance, not bit-for-bit across builds of the same edge list; the
        CPU path's sorted CSR is the bitwise-reproducible oracle."""
        if src.is_cuda:
            from . import ops
            rowptr, tails, weights = ops.csr_from_edges(nv, base, src, dst, w)
            return Graph(rowptr, tails, weights)
        lsrc = src - base
        # stable two-pass lexicographic sort: by dst then by src
        o1 = torch.argsort(dst, stable=True)
        lsrc1, dst1, w1 = lsrc[o1], dst[o1], w[o1]
        o2 = torch.argsort(lsrc1, stable=True)
        lsrc2, dst2, w2 = lsrc1[o2], dst1[o2], w1[o2]
        rowptr = torch.zeros(nv + 1, dtype=torch.int64, device=src.device)
        if lsrc2.numel():
            rowptr[1:] = torch.cumsum(torch.bincount(lsrc2, minlength=nv), dim=0)
        return Graph(rowptr, dst2, w2)


class DistGraph:
    """A rank's shard of a distributed graph: the local CSR + the partition.

    Reference: DistGraph (distgraph.hpp:27-237). Ghost discovery is done per
    phase by the halo engine (halo.py), not stored here.
    """

    def __init__(self, graph: Graph, partition: Partition, rank: int):
        assert graph.nv == partition.nv_local(rank)
        self.g = graph
        self.partition = partition
        self.rank = rank

    @property
    def base(self) -> int:
        return self.partition.base(self.rank)

    @property
    def bound(self) -> int:
        return self.partition.bound(self.rank)

    @property
    def nv(self) -> int:
        return self.g.nv

    @property
    def nv_global(self) -> int:
        return self.partition.nv_global

    @property
    def ne(self) -> int:
        return self.g.ne

    def to(self, device) -> "DistGraph":
        return DistGraph(self.g.to(device), self.partition, self.rank)

    def ghost_vertices(self) -> torch.Tensor:
        """Sorted unique global ids of remote tails (the rank's ghosts).
        Chunked: torch advanced indexing overflows internal int32 offsets
        past ~2^31 elements (observed as an absurd-size alloc at R-MAT s26)."""
        if self.g._ne_released >= 0:
            raise RuntimeError("tails released: build halos before "
                               "Graph.release_tails()")
        t = self.g.tails
        if self.partition.nranks == 1:
            return torch.empty(0, dtype=torch.int64, device=t.device)
        base, bound = self.base, self.bound
        CH = 1 << 28
        parts = []
        for c0 in range(0, t.numel(), CH):
            tc = t[c0:c0 + CH]
            parts.append(torch.unique(tc[(tc < base) | (tc >= bound)]))
        if not parts:
            return torch.empty(0, dtype=torch.int64, device=t.device)
        return torch.unique(torch.cat(parts))

    def local_degree_sum(self) -> torch.Tensor:
        """Per-vertex weighted degree (vDegree; ref louvain.cpp:2126-2151)."""
        nv = self.nv
        # loud guard: an emptied/mismatched tails array would silently zero
        # every degree here (one 8-byte D2H read, once per phase)
        assert self.g.ne == int(self.g.rowptr[-1]), \
            "CSR tails/rowptr inconsistent (tails freed or truncated?)"
        if self.g.ne == 0:
            return torch.zeros(nv, dtype=self.g.weights.dtype, device=self.g.device)
        if self.g.device.type == "cuda":
            from . import ops
            return ops.row_sum(self.g.rowptr, self.g.weights)
        seg = torch.repeat_interleave(
            torch.arange(nv, device=self.g.device), self.g.degrees()
        )
        out = torch.zeros(nv, dtype=self.g.weights.dtype, device=self.g.device)
        out.index_add_(0, seg, self.g.weights)
        return out


def single_partition(g: Graph) -> DistGraph:
    """Wrap a whole graph as a world-size-1 DistGraph."""
    return DistGraph(g, Partition.contiguous(g.nv, 1), 0)
