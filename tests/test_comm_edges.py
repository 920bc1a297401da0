"""Hardening tests for the communication layer ahead of its first RCCL/xGMI
hardware contact (round-1 verdict item 5): every collective call site
exercised with empty, asymmetric and large payloads on gloo (the identical
code path runs over nccl/RCCL on the MI355X node)."""

import pytest
import torch

from tests.dist_utils import run_dist

from cuvite_amd.parallel import Comm


# ---- workers ---------------------------------------------------------------

def _w_a2av_all_empty(rank, world):
    comm = Comm()
    send = [torch.empty(0, dtype=torch.int64) for _ in range(world)]
    got = comm.all_to_all_v(send)
    return all(g.numel() == 0 for g in got)


def _w_a2av_asymmetric(rank, world):
    """Rank r sends (r+1)*10+p elements to peer p only when r is even;
    odd ranks send nothing. Every pairing of empty/nonempty send+recv."""
    comm = Comm()
    if rank % 2 == 0:
        send = [torch.full(((rank + 1) * 10 + p,), rank * 100 + p,
                           dtype=torch.int64) for p in range(world)]
    else:
        send = [torch.empty(0, dtype=torch.int64) for _ in range(world)]
    got = comm.all_to_all_v(send)
    ok = True
    for p in range(world):
        if p % 2 == 0:
            exp_n = (p + 1) * 10 + rank
            ok &= got[p].numel() == exp_n
            ok &= bool((got[p] == p * 100 + rank).all())
        else:
            ok &= got[p].numel() == 0
    return ok


def _w_a2av_large(rank, world):
    """~1M fp64 elements per pair (payloads the halo exchange ships at s26)."""
    comm = Comm()
    n = 1 << 20
    send = [torch.full((n,), float(rank * world + p), dtype=torch.float64)
            for p in range(world)]
    got = comm.all_to_all_v(send)
    return all(got[p].numel() == n and float(got[p][0]) == p * world + rank
               and float(got[p][-1]) == p * world + rank
               for p in range(world))


def _w_a2av_precounted(rank, world):
    """recv_counts supplied (the fixed-size reply path) incl. zero counts."""
    comm = Comm()
    send = [torch.arange(p, dtype=torch.float64) for p in range(world)]
    # peer p sends me `rank` elements
    got = comm.all_to_all_v(send, recv_counts=[rank] * world)
    return all(got[p].numel() == rank for p in range(world))


def _w_exchange_fixed_mixed(rank, world):
    """Pre-negotiated exchange with a mix of zero and nonzero buffers."""
    comm = Comm()
    # rank r sends r elements to every peer
    send = [torch.full((rank,), float(rank), dtype=torch.float64)
            for _ in range(world)]
    recv = [torch.empty(p, dtype=torch.float64) for p in range(world)]
    comm.exchange_fixed(send, recv)
    ok = True
    for p in range(world):
        if p != rank:
            ok &= bool((recv[p] == float(p)).all())
    return ok


def _w_gather_cat(rank, world):
    comm = Comm()
    # variable lengths incl. an empty rank
    t = torch.arange(rank * 3, dtype=torch.int64) + rank * 1000
    got = comm.gather_cat(t, root=0)
    if rank != 0:
        return got is None
    exp = torch.cat([torch.arange(r * 3, dtype=torch.int64) + r * 1000
                     for r in range(world)])
    return bool(torch.equal(got, exp))


def _w_fetch_info_lists_empty(rank, world):
    """fetch_comm_info_lists with all-empty requests still completes
    collectively (a rank with no referenced remote communities must not
    deadlock its peers)."""
    from cuvite_amd.generators import karate_graph
    from cuvite_amd.graph import Graph, DistGraph, Partition
    from cuvite_amd.halo import build_halo, fetch_comm_info_lists
    g = karate_graph()
    part = Partition.contiguous(g.nv, world)
    b, e = part.base(rank), part.bound(rank)
    rp = g.rowptr[b:e + 1] - g.rowptr[b]
    e0, e1 = int(g.rowptr[b]), int(g.rowptr[e])
    dg = DistGraph(Graph(rp.clone(), g.tails[e0:e1].clone(),
                         g.weights[e0:e1].clone()), part, rank)
    comm = Comm()
    ctx = build_halo(dg, comm)
    size = torch.ones(dg.nv, dtype=torch.int64)
    degree = torch.ones(dg.nv, dtype=torch.float64)
    if rank == 0:
        # rank 0 requests one community from each other owner
        reqs = [torch.empty(0, dtype=torch.int64)] + \
            [torch.tensor([int(part.parts[p])], dtype=torch.int64)
             for p in range(1, world)]
    else:
        reqs = [torch.empty(0, dtype=torch.int64) for _ in range(world)]
    sizes, degrees = fetch_comm_info_lists(ctx, reqs, size, degree)
    if rank == 0:
        return all(sizes[p].numel() == 1 and int(sizes[p][0]) == 1
                   for p in range(1, world))
    return all(s.numel() == 0 for s in sizes)


def _w_allreduce_scalar(rank, world):
    comm = Comm()
    s = comm.allreduce_scalar(float(rank + 1))
    m = comm.allreduce_scalar(float(rank), op="max")
    return s == world * (world + 1) / 2 and m == world - 1


# ---- tests -----------------------------------------------------------------

@pytest.mark.parametrize("world", [2, 3])
def test_all_to_all_v_all_empty(world):
    assert all(run_dist(world, _w_a2av_all_empty))


@pytest.mark.parametrize("world", [2, 4])
def test_all_to_all_v_asymmetric_empties(world):
    assert all(run_dist(world, _w_a2av_asymmetric))


def test_all_to_all_v_large_payload():
    assert all(run_dist(2, _w_a2av_large))


def test_all_to_all_v_precounted_zero_counts():
    assert all(run_dist(3, _w_a2av_precounted))


def test_exchange_fixed_mixed_sizes():
    assert all(run_dist(3, _w_exchange_fixed_mixed))


@pytest.mark.parametrize("world", [2, 4])
def test_gather_cat_variable_with_empty(world):
    assert all(run_dist(world, _w_gather_cat))


def test_fetch_comm_info_lists_empty_requests():
    assert all(run_dist(3, _w_fetch_info_lists_empty))


def test_allreduce_scalars():
    assert all(run_dist(3, _w_allreduce_scalar))
