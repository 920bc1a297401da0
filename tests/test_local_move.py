import numpy as np
import pytest
import torch

from cuvite_amd.generators import karate_graph, rmat_graph
from cuvite_amd.graph import single_partition
from cuvite_amd.local_move import (MoveInputs, local_move_pydict,
                                   local_move_torch, modularity_parts)


def _singleton_inputs(g):
    dg = single_partition(g)
    vdeg = dg.local_degree_sum()
    nv = g.nv
    return MoveInputs(
        rowptr=g.rowptr,
        tails=g.tails.to(torch.int32),
        weights=g.weights,
        curr_comm=torch.arange(nv, dtype=torch.int32),
        v_degree=vdeg,
        comm_size=torch.ones(nv, dtype=torch.int64),
        comm_degree=vdeg.clone(),
        comm_gid=torch.arange(nv, dtype=torch.int64),
        constant=1.0 / float(vdeg.to(torch.float64).sum()),
    )


def _random_comm_inputs(g, seed):
    rng = torch.Generator().manual_seed(seed)
    nv = g.nv
    dg = single_partition(g)
    vdeg = dg.local_degree_sum()
    curr = torch.randint(0, nv, (nv,), generator=rng)
    # aggregates consistent with the labels
    size = torch.zeros(nv, dtype=torch.int64)
    size.index_add_(0, curr, torch.ones(nv, dtype=torch.int64))
    cdeg = torch.zeros(nv, dtype=vdeg.dtype)
    cdeg.index_add_(0, curr, vdeg)
    return MoveInputs(g.rowptr, g.tails.to(torch.int32), g.weights,
                      curr.to(torch.int32), vdeg, size, cdeg,
                      torch.arange(nv, dtype=torch.int64),
                      1.0 / float(vdeg.to(torch.float64).sum()))


@pytest.mark.parametrize("case", ["karate_singleton", "karate_random",
                                  "rmat_singleton", "rmat_random"])
def test_torch_matches_pydict(case):
    if case.startswith("karate"):
        g = karate_graph()
    else:
        g = rmat_graph(7, 8, seed=11)
    if case.endswith("singleton"):
        inp = _singleton_inputs(g)
    else:
        inp = _random_comm_inputs(g, seed=23)
    t_ref, cw_ref = local_move_pydict(inp)
    t_vec, cw_vec = local_move_torch(inp)
    assert torch.equal(t_ref.to(torch.int64), t_vec.to(torch.int64))
    assert torch.allclose(cw_ref, cw_vec, atol=1e-12)


def test_first_iteration_karate_moves():
    g = karate_graph()
    inp = _singleton_inputs(g)
    target, cw = local_move_torch(inp)
    # from singletons, no weight to own community yet
    assert float(cw.sum()) == 0.0
    # singleton guard: vertices only move to LOWER ids on the first sweep
    t = target.to(torch.int64)
    moved = t != torch.arange(34)
    assert bool(moved.any())
    assert bool((t[moved] < torch.arange(34)[moved]).all())


def test_isolated_vertex_stays():
    rowptr = torch.tensor([0, 2, 3, 3, 4], dtype=torch.int64)
    tails = torch.tensor([1, 3, 0, 0], dtype=torch.int32)
    w = torch.ones(4, dtype=torch.float64)
    inp = MoveInputs(rowptr, tails, w,
                     torch.arange(4, dtype=torch.int32),
                     torch.tensor([2.0, 1.0, 0.0, 1.0], dtype=torch.float64),
                     torch.ones(4, dtype=torch.int64),
                     torch.tensor([2.0, 1.0, 0.0, 1.0], dtype=torch.float64),
                     torch.arange(4, dtype=torch.int64),
                     0.25)
    t, cw = local_move_torch(inp)
    assert int(t[2]) == 2
    tr, cwr = local_move_pydict(inp)
    assert torch.equal(t.to(torch.int64), tr.to(torch.int64))


def test_self_loop_semantics():
    # vertex 0 has a self loop; it contributes to clusterWeight but not to eix
    rowptr = torch.tensor([0, 2, 3], dtype=torch.int64)
    tails = torch.tensor([0, 1, 0], dtype=torch.int32)
    w = torch.tensor([5.0, 1.0, 1.0], dtype=torch.float64)
    vdeg = torch.tensor([6.0, 1.0], dtype=torch.float64)
    inp = MoveInputs(rowptr, tails, w,
                     torch.arange(2, dtype=torch.int32), vdeg,
                     torch.ones(2, dtype=torch.int64), vdeg.clone(),
                     torch.arange(2, dtype=torch.int64),
                     1.0 / 7.0)
    t, cw = local_move_torch(inp)
    tr, cwr = local_move_pydict(inp)
    assert torch.equal(t.to(torch.int64), tr.to(torch.int64))
    assert torch.allclose(cw, cwr)
    assert float(cw[0]) == 5.0  # self-loop counted in counter[cc]


def test_modularity_parts():
    cw = torch.tensor([1.0, 2.0], dtype=torch.float64)
    cd = torch.tensor([3.0, 4.0], dtype=torch.float64)
    p = modularity_parts(cw, cd)
    assert p.tolist() == [3.0, 25.0]


def test_hub_moves_sorted_matches_oracle():
    """The sort-based hub path (ops._hub_moves_sorted) must reproduce the
    torch oracle exactly on unit-weight graphs, including tie-breaks and the
    singleton guard. Runs on CPU (pure torch ops)."""
    from cuvite_amd.generators import rmat_graph
    from cuvite_amd import ops

    torch.manual_seed(3)
    g = rmat_graph(9, 24, seed=7)
    g.weights.fill_(1.0)
    nv = g.nv
    for state in ("singleton", "random"):
        if state == "singleton":
            curr = torch.arange(nv, dtype=torch.int64)
        else:
            curr = torch.randint(0, nv, (nv,),
                                 generator=torch.Generator().manual_seed(4))
        size = torch.zeros(nv, dtype=torch.int64)
        size.index_add_(0, curr, torch.ones(nv, dtype=torch.int64))
        deg = g.rowptr[1:] - g.rowptr[:-1]
        vdeg = torch.zeros(nv, dtype=torch.float64)
        seg = torch.repeat_interleave(torch.arange(nv), deg)
        vdeg.index_add_(0, seg, g.weights)
        cdeg = torch.zeros(nv, dtype=torch.float64)
        cdeg.index_add_(0, curr, vdeg)
        inp = MoveInputs(g.rowptr, g.tails.to(torch.int32), g.weights,
                         curr.to(torch.int32), vdeg, size, cdeg,
                         torch.arange(nv, dtype=torch.int64),
                         1.0 / float(vdeg.sum()))
        t_ref, cw_ref = local_move_torch(inp)
        # treat every vertex with deg > 64 as a "hub" for this test
        hubs = (deg > 64).nonzero(as_tuple=True)[0]
        assert hubs.numel() > 3, "need hubs for the test to be meaningful"
        hub_tgt, hub_cw = ops._hub_moves_sorted(inp, hubs, deg[hubs])
        assert torch.equal(hub_tgt.to(torch.int64),
                           t_ref[hubs].to(torch.int64))
        assert torch.allclose(hub_cw, cw_ref[hubs])


def test_hub_moves_sorted_chunked_matches_unchunked():
    from cuvite_amd.generators import rmat_graph
    from cuvite_amd import ops

    g = rmat_graph(8, 24, seed=2)
    nv = g.nv
    deg = g.rowptr[1:] - g.rowptr[:-1]
    curr = torch.arange(nv, dtype=torch.int64)
    vdeg = torch.zeros(nv, dtype=torch.float64)
    seg = torch.repeat_interleave(torch.arange(nv), deg)
    vdeg.index_add_(0, seg, g.weights)
    inp = MoveInputs(g.rowptr, g.tails.to(torch.int32), g.weights,
                     curr.to(torch.int32), vdeg,
                     torch.ones(nv, dtype=torch.int64), vdeg.clone(),
                     torch.arange(nv, dtype=torch.int64),
                     1.0 / float(vdeg.sum()))
    hubs = (deg > 32).nonzero(as_tuple=True)[0]
    t1, cw1 = ops._hub_moves_sorted(inp, hubs, deg[hubs])
    old = ops._HUB_SORT_CHUNK
    try:
        ops._HUB_SORT_CHUNK = 200  # force many groups
        t2, cw2 = ops._hub_moves_sorted(inp, hubs, deg[hubs])
    finally:
        ops._HUB_SORT_CHUNK = old
    assert torch.equal(t1, t2)
    assert torch.allclose(cw1, cw2)


def test_fuzz_oracle_vs_pydict():
    """Randomized graphs + random community states: the vectorized oracle and
    the dict-based golden transcription must agree exactly."""
    for seed in range(12):
        g = torch.Generator().manual_seed(900 + seed)
        nv = int(torch.randint(2, 24, (1,), generator=g))
        ne = int(torch.randint(1, 80, (1,), generator=g))
        src = torch.randint(0, nv, (ne,), generator=g)
        dst = torch.randint(0, nv, (ne,), generator=g)
        # unit weights: exact arithmetic in any order
        w = torch.ones(ne, dtype=torch.float64)
        from cuvite_amd.graph import Graph
        gr = Graph.from_edge_tuples(nv, torch.cat([src, dst]),
                                    torch.cat([dst, src]),
                                    torch.cat([w, w]))
        curr = torch.randint(0, nv, (nv,), generator=g)
        size = torch.zeros(nv, dtype=torch.int64)
        size.index_add_(0, curr, torch.ones(nv, dtype=torch.int64))
        deg = gr.rowptr[1:] - gr.rowptr[:-1]
        vdeg = torch.zeros(nv, dtype=torch.float64)
        seg = torch.repeat_interleave(torch.arange(nv), deg)
        vdeg.index_add_(0, seg, gr.weights)
        cdeg = torch.zeros(nv, dtype=torch.float64)
        cdeg.index_add_(0, curr, vdeg)
        tot = float(vdeg.sum())
        inp = MoveInputs(gr.rowptr, gr.tails.to(torch.int32), gr.weights,
                         curr.to(torch.int32), vdeg, size, cdeg,
                         torch.arange(nv, dtype=torch.int64),
                         1.0 / tot if tot else 0.0)
        t1, c1 = local_move_torch(inp)
        t2, c2 = local_move_pydict(inp)
        assert torch.equal(t1.to(torch.int64), t2.to(torch.int64)), seed
        assert torch.allclose(c1, c2), seed
