"""Graph coarsening (phase transition): collapse communities into vertices.

Reference: distReNumber + fill_newEdgesMap + send_newEdges
(rebuild.cpp:27-454). MI355X-native redesign: the reference renumbers through
std::map and aggregates edges in nested maps on the host, serially; here every
step is a device tensor op (sort / unique / searchsorted / segment sums), and
edges are pre-aggregated locally before the exchange so the all-to-all traffic
is the coarse edge list, not the fine one.

Steps:
  1. Surviving communities are routed to their owners (owner of community c =
     owner of vertex id c); each owner assigns dense new ids to its own
     survivors; a count allgather turns them into global new ids (contiguous
     in rank order, so the new 1-D partition is exactly "what each owner
     numbered"), matching distReNumber's offsets (rebuild.cpp:181).
  2. Every rank resolves old-comm -> new-id for all communities it references
     (its own labels + its ghosts' labels) with one owner lookup.
  3. Edge tuples (new_src, new_dst, w) are aggregated locally, routed to the
     new owner of new_src, merged again, and assembled into the new CSR.
"""

from __future__ import annotations

from typing import Callable, Tuple

import torch

from .graph import DistGraph, Graph, Partition
from .halo import build_halo, exchange_ghost_labels
from .parallel import Comm


def _owner_lookup(comm: Comm, part: Partition, query: torch.Tensor,
                  reply_fn) -> torch.Tensor:
    """Generic owner-based lookup: route sorted `query` gids to owners, each
    owner answers via reply_fn(local_ids) -> values, results return aligned
    with `query` (which must be sorted)."""
    dev = query.device
    if comm.world == 1:
        return reply_fn(query)
    parts = part.parts.to(dev)
    offs = torch.searchsorted(query, parts)
    reqs = [query[offs[p]:offs[p + 1]] for p in range(comm.world)]
    got = comm.all_to_all_v(reqs)
    replies = [reply_fn(g) if g.numel() else
               torch.empty(0, dtype=torch.int64, device=dev) for g in got]
    back = comm.all_to_all_v(replies, recv_counts=[int(r.numel()) for r in reqs])
    return torch.cat([back[p] for p in range(comm.world)])


def coarsen(dg: DistGraph, comm: Comm, cvect: torch.Tensor,
            halo=None
            ) -> Tuple[DistGraph, Callable[[torch.Tensor], torch.Tensor]]:
    """Build the next-level graph from community labels `cvect` (int64 [nv],
    global old-community ids). Returns (new DistGraph for this rank, renum)
    where renum maps any tensor of old community gids to new vertex gids
    (collective: all ranks must call it together with their own queries).
    `halo`: reuse the phase's HaloContext (ghost set is label-independent,
    so the rebuild is redundant)."""
    dev = dg.g.device
    base, bound = dg.base, dg.bound
    world, rank = comm.world, comm.rank

    # --- 1. survivors to owners; owners number their own ---------------------
    my_labels = torch.unique(cvect)  # sorted
    if world == 1:
        survivors = my_labels
    else:
        parts = dg.partition.parts.to(dev)
        offs = torch.searchsorted(my_labels, parts)
        reqs = [my_labels[offs[p]:offs[p + 1]] for p in range(world)]
        got = comm.all_to_all_v(reqs)
        survivors = torch.unique(torch.cat(got))  # sorted, all owned by me
    lnc = survivors.numel()
    counts = torch.zeros(world, dtype=torch.int64, device=comm.device)
    counts[rank] = lnc
    counts = comm.allgather_counts(counts).sum(dim=0) if world > 1 else counts
    new_offsets = torch.zeros(world + 1, dtype=torch.int64)
    new_offsets[1:] = torch.cumsum(counts.cpu(), dim=0)
    my_off = int(new_offsets[rank])
    gnc = int(new_offsets[-1])
    assert gnc < (1 << 31), "coarse graph must have < 2^31 communities"
    new_part = Partition.from_bounds(new_offsets)

    def reply_newid(gids: torch.Tensor) -> torch.Tensor:
        return my_off + torch.searchsorted(survivors, gids)

    def renum(gids: torch.Tensor) -> torch.Tensor:
        flat_sorted, order = torch.sort(gids)
        vals = _owner_lookup(comm, dg.partition, flat_sorted, reply_newid)
        out = torch.empty_like(vals)
        out[order] = vals
        return out

    # --- 2. resolve new ids for all referenced communities -------------------
    if halo is None:
        halo = build_halo(dg, comm)
    ghost_cvect = exchange_ghost_labels(halo, cvect)
    all_cvect = torch.cat([cvect, ghost_cvect])
    ref_uniq = torch.unique(all_cvect)  # sorted
    ref_new = _owner_lookup(comm, dg.partition, ref_uniq, reply_newid)
    # map labels via position in ref_uniq
    new_of_all = ref_new[torch.searchsorted(ref_uniq, all_cvect)]

    # --- 3. coarse edges: aggregate locally, route, merge, CSR ---------------
    # chunked over the edge list: torch sort/advanced-indexing are capped at
    # INT_MAX elements and the per-chunk aggregates are far smaller than the
    # fine edge list, so partial aggregation also bounds peak memory
    # coarse edges live as packed keys s*gnc+t (16 B/entry with the weight
    # instead of 24 — the s component is never materialized: routing, the
    # CSR rowptr and the tails all derive from the key directly)
    ne = dg.g.ne
    CH = 1 << 28
    rowptr = dg.g.rowptr
    W = dg.g.weights.dtype
    parts_k, parts_w = [], []

    for c0 in range(0, max(ne, 1), CH):
        c1 = min(c0 + CH, ne)
        if c1 <= c0:
            break
        eidx = torch.arange(c0, c1, device=dev)
        seg = torch.searchsorted(rowptr, eidx, right=True) - 1
        del eidx
        key = new_of_all[seg] * gnc
        del seg
        key += new_of_all[halo.tails_dense[c0:c1].to(torch.int64)]
        ck, cw = _aggregate_keys(key, dg.g.weights[c0:c1])
        del key
        parts_k.append(ck)
        parts_w.append(cw)
    k_agg, w_agg = _fold_keys(parts_k, parts_w, dev, W)
    del parts_k, parts_w

    if world > 1:
        npdev = new_part.parts.to(dev)
        offs = torch.searchsorted(k_agg, npdev * gnc)
        got_k = comm.all_to_all_v(
            [k_agg[offs[p]:offs[p + 1]] for p in range(world)])
        cnts = [int(g.numel()) for g in got_k]
        got_w = comm.all_to_all_v(
            [w_agg[offs[p]:offs[p + 1]] for p in range(world)],
            recv_counts=cnts)
        # each peer's contribution arrives key-sorted: range-fold them
        # (one flat sort would break the INT_MAX cap at s30-per-rank sizes)
        k_agg, w_agg = _fold_keys(got_k, got_w, dev, W)

    nbase = new_part.base(rank)
    nv_new = new_part.nv_local(rank)
    if k_agg.numel():
        # key-sorted, so the CSR rowptr is a direct searchsorted
        # (bincount is capped at INT_MAX input elements; this is not)
        rowptr = torch.searchsorted(
            k_agg, torch.arange(nbase, nbase + nv_new + 1,
                                device=dev) * gnc)
        t_agg = k_agg % gnc
    else:
        rowptr = torch.zeros(nv_new + 1, dtype=torch.int64, device=dev)
        t_agg = torch.zeros(0, dtype=torch.int64, device=dev)
    del k_agg
    new_g = Graph(rowptr, t_agg, w_agg)
    return DistGraph(new_g, new_part, rank), renum


# one merge sort must stay under both the torch sort cap (INT_MAX elements)
# and a sane transient-memory footprint (the cat + rocPRIM double buffers of
# a 1.6G-entry fold were ~80 GB and OOM'd the s27 converged run)
_FOLD_CAP = 1 << 29


def _fold_keys(parts_k, parts_w, dev, W):
    """Merge key-sorted (key, weight) chunk aggregates into one globally
    aggregated, key-sorted pair. Small totals: one cat + aggregate. Large
    totals (s27/s30-per-rank): partition the KEY SPACE into ranges from the
    largest part's quantiles and aggregate each range independently — no
    single sort exceeds _FOLD_CAP elements and the total result may exceed
    2^31 entries (only ever touched by elementwise/searchsorted ops after
    this). A key appears at most once per part, so ranges can always be
    split below the cap."""
    parts_k = [p for p in parts_k if p.numel()]
    parts_w = [p for p in parts_w if p.numel()]
    if not parts_k:
        return (torch.zeros(0, dtype=torch.int64, device=dev),
                torch.zeros(0, dtype=W, device=dev))
    if len(parts_k) == 1:
        return parts_k[0], parts_w[0]
    total = sum(int(p.numel()) for p in parts_k)
    if total <= _FOLD_CAP:
        return _aggregate_keys(torch.cat(parts_k), torch.cat(parts_w))
    big = max(range(len(parts_k)), key=lambda i: parts_k[i].numel())
    n_ranges = (total + _FOLD_CAP // 2 - 1) // (_FOLD_CAP // 2)
    bk = parts_k[big]
    qpos = torch.linspace(0, bk.numel() - 1, n_ranges + 1,
                          device=dev).to(torch.int64)[1:-1]
    bounds = [None] + [int(bk[q]) for q in qpos] + [None]
    out_k, out_w = [], []
    for r in range(len(bounds) - 1):
        lo, hi = bounds[r], bounds[r + 1]
        sl_k, sl_w = [], []
        for pk, pw in zip(parts_k, parts_w):
            a = 0 if lo is None else int(torch.searchsorted(
                pk, torch.tensor(lo, device=dev)))
            b = pk.numel() if hi is None else int(torch.searchsorted(
                pk, torch.tensor(hi, device=dev)))
            if b > a:
                sl_k.append(pk[a:b])
                sl_w.append(pw[a:b])
        if not sl_k:
            continue
        n_r = sum(int(x.numel()) for x in sl_k)
        if len(sl_k) == 1:
            # single sorted, already-deduped slice: pass through
            ck, cw = sl_k[0], sl_w[0]
        elif n_r > _FOLD_CAP:
            # rare skewed range: recurse with the slice list
            ck, cw = _fold_keys(sl_k, sl_w, dev, W)
        else:
            ck, cw = _aggregate_keys(torch.cat(sl_k), torch.cat(sl_w))
        out_k.append(ck)
        out_w.append(cw)
    return torch.cat(out_k), torch.cat(out_w)


def _aggregate_keys(key: torch.Tensor, w: torch.Tensor):
    """Sum weights of duplicate keys; returns (uniq keys sorted, sums),
    both exactly m-sized (full-buffer views are CLONED — a [:m] slice of the
    rocPRIM output would pin the n-sized buffer, which leaked 64 GB across
    the s27 chunk aggregates before this was a clone)."""
    if key.is_cuda:
        from . import ops
        if ops.available():
            # narrow-bit rocPRIM radix sort (only the bits the key uses,
            # ~50 at s26 vs 64 for a generic int64 sort) + reduce_by_key —
            # replaces the sort / gather / unique_consecutive /
            # double-cumsum chain that dominated the 5.9 s phase-0 rebuild
            end_bit = int(key.max()).bit_length() if key.numel() else 1
            w64 = w if w.dtype == torch.float64 else w.to(torch.float64)
            uniq, sums, cnt = ops._require().sort_reduce_pairs(
                key, w64, max(1, min(end_bit, 64)))
            m = int(cnt[0])
            return uniq[:m].clone(), sums[:m].to(w.dtype).clone()
    key_s, order = torch.sort(key)
    w_s = w[order]
    uniq, counts = torch.unique_consecutive(key_s, return_counts=True)
    # segmented sum over sorted runs via cumsum (no atomics: torch fp64
    # scatter-adds are CAS loops on ROCm, and this is deterministic)
    ends = torch.cumsum(counts, dim=0) - 1
    cs = torch.cumsum(w_s.to(torch.float64), dim=0)
    w_out = cs[ends].clone()
    w_out[1:] -= cs[ends[:-1]]
    return uniq, w_out.to(w.dtype)


def _aggregate(s: torch.Tensor, t: torch.Tensor, w: torch.Tensor, gnc: int):
    """Merge duplicate (s, t) directed edges, summing weights; returns sorted
    by (s, t). Keys fit int64 because gnc < 2^31."""
    uniq, sums = _aggregate_keys(s * gnc + t, w)
    return uniq // gnc, uniq % gnc, sums


def remap_labels(dg: DistGraph, comm: Comm, assign: torch.Tensor,
                 cvect: torch.Tensor) -> torch.Tensor:
    """Compose clusterings across a phase: `assign[j]` is the current-level
    vertex gid that original vertex j belongs to; return cvect[assign[j]]
    fetched from the owning ranks (distributed version of the reference's
    root-side commAll[p] = cvectAll[commAll[p]], main.cpp:397-404)."""
    def reply(gids: torch.Tensor) -> torch.Tensor:
        return cvect[gids - dg.base]

    flat_sorted, order = torch.sort(assign)
    vals = _owner_lookup(comm, dg.partition, flat_sorted, reply)
    out = torch.empty_like(vals)
    out[order] = vals
    return out
