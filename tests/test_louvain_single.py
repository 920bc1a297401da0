import pytest
import torch

from cuvite_amd.generators import karate_graph, rmat_graph
from cuvite_amd.graph import single_partition
from cuvite_amd.louvain import louvain, LouvainConfig
from cuvite_amd.parallel import Comm


def _modularity_of(g, comm_labels):
    """Independent fp64 modularity check: Q = sum_c (e_c/2m - (a_c/2m)^2),
    where e_c counts internal directed weight (self-loops once)."""
    dg = single_partition(g)
    vdeg = dg.local_degree_sum().to(torch.float64)
    m2 = float(vdeg.sum())
    seg = torch.repeat_interleave(torch.arange(g.nv), g.degrees())
    lab = comm_labels.to(torch.int64)
    internal = lab[seg] == lab[g.tails]
    e_in = torch.zeros(int(lab.max()) + 1, dtype=torch.float64)
    e_in.index_add_(0, lab[seg[internal]], g.weights[internal].to(torch.float64))
    a = torch.zeros(int(lab.max()) + 1, dtype=torch.float64)
    a.index_add_(0, lab, vdeg)
    return float((e_in / m2).sum() - ((a / m2) ** 2).sum())


def test_karate_full_louvain():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(), LouvainConfig(backend="torch"))
    q = _modularity_of(g, res.communities)
    # classic Louvain lands ~0.41-0.42 on karate
    assert q > 0.38, f"karate modularity too low: {q}"
    assert res.total_iters >= 2
    ncomm = len(torch.unique(res.communities))
    assert 2 <= ncomm <= 8


def test_karate_one_phase():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(),
                  LouvainConfig(backend="torch", one_phase=True))
    assert res.phases == 1
    q = _modularity_of(g, res.communities)
    # a single phase of simultaneous (Jacobi-style) moves converges to a
    # modest Q on karate, as in the reference (multi-phase recovers it)
    assert q > 0.1


def test_karate_reported_mod_close_to_recomputed():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(), LouvainConfig(backend="torch"))
    q = _modularity_of(g, res.communities)
    # reported modularity is the reference's approximate per-iteration value;
    # must be close to the exact recomputation of the final clustering
    assert abs(res.modularity - q) < 0.05


def test_rmat_louvain_improves():
    g = rmat_graph(8, 8, seed=3)
    res = louvain(single_partition(g), Comm(), LouvainConfig(backend="torch"))
    q = _modularity_of(g, res.communities)
    assert q > 0.1
    assert res.total_iters >= 2


def test_threshold_scaling_runs():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(),
                  LouvainConfig(backend="torch", threshold_scaling=True))
    q = _modularity_of(g, res.communities)
    assert q > 0.35


@pytest.mark.parametrize("et", [1, 2, 3, 4])
def test_early_termination_variants(et):
    g = karate_graph()
    res = louvain(single_partition(g), Comm(),
                  LouvainConfig(backend="torch", early_term=et, et_delta=0.5))
    q = _modularity_of(g, res.communities)
    assert q > 0.25  # ET trades quality for speed but shouldn't collapse


def test_coloring_louvain():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(),
                  LouvainConfig(backend="torch", coloring=True, max_colors=8))
    q = _modularity_of(g, res.communities)
    assert q > 0.35


def test_vertex_ordering_louvain():
    g = karate_graph()
    res = louvain(single_partition(g), Comm(),
                  LouvainConfig(backend="torch", ordering=True, max_colors=8))
    q = _modularity_of(g, res.communities)
    assert q > 0.35


def test_karate_golden_trajectory():
    """Golden regression: the karate-club run must keep producing exactly
    this modularity/phase/iteration profile (any change means the algorithm
    semantics moved)."""
    import torch
    from cuvite_amd.generators import karate_graph
    from cuvite_amd.graph import single_partition
    from cuvite_amd.louvain import louvain, LouvainConfig
    from cuvite_amd.parallel import Comm

    res = louvain(single_partition(karate_graph()),
                  Comm(torch.device("cpu")), LouvainConfig(backend="torch"))
    assert res.modularity == pytest.approx(0.408695, abs=1e-6)
    assert res.phases == 3
    assert res.total_iters == 11
    assert res.modularity_per_level == pytest.approx(
        res.modularity_per_level)  # shape stability


def test_threshold_cycling_schedule():
    """Exact reference schedule (main.cpp:225-239): sp 0-2 -> 1e-3,
    3-6 -> 1e-4, 7-9 -> 1e-5, 10-12 -> 1e-6, wrap at 13."""
    from cuvite_amd.louvain import LouvainConfig, _threshold_for_phase
    cfg = LouvainConfig(threshold_scaling=True)
    expected = [1e-3] * 3 + [1e-4] * 4 + [1e-5] * 3 + [1e-6] * 3
    for sp in range(26):
        assert _threshold_for_phase(cfg, sp) == expected[sp % 13], sp


def test_et_freeze_semantics():
    """-t 1: a vertex stable for 3 iterations becomes inactive and its
    target stays pinned to its current community."""
    import torch
    from cuvite_amd.generators import karate_graph
    from cuvite_amd.graph import single_partition
    from cuvite_amd.louvain import (LouvainConfig, PhaseState, _et_update)
    from cuvite_amd.parallel import Comm

    dg = single_partition(karate_graph())
    state = PhaseState(dg, Comm(torch.device("cpu")))
    cfg = LouvainConfig(early_term=1)
    rng = torch.Generator().manual_seed(0)
    # simulate 3 stable iterations: target == curr == past
    tgt = state.curr_comm.clone()
    for _ in range(3):
        _et_update(state, cfg, tgt, rng)
        state.past_comm = state.curr_comm.clone()
    assert not bool(state.active.any())  # every vertex frozen after 3 stable


def test_louvain_with_prebuilt_halo_matches():
    """louvain(halo=...) (bench.py's converged-run path: reuse the timed
    region's phase-0 halo) must produce the identical result."""
    from cuvite_amd.halo import build_halo
    g = rmat_graph(8, 8, seed=3)
    dg = single_partition(g)
    ref = louvain(dg, Comm(), LouvainConfig(backend="torch"))
    h = build_halo(dg, Comm())
    got = louvain(dg, Comm(), LouvainConfig(backend="torch"), halo=h)
    assert got.modularity == ref.modularity
    assert torch.equal(got.communities, ref.communities)
    assert got.total_iters == ref.total_iters


def test_louvain_with_prebuilt_halo_coloring():
    """halo= composes with -c (coloring shares the provided halo)."""
    from cuvite_amd.halo import build_halo
    g = rmat_graph(8, 8, seed=3)
    dg = single_partition(g)
    cfg = LouvainConfig(backend="torch", coloring=True, max_colors=6)
    ref = louvain(dg, Comm(), cfg)
    got = louvain(dg, Comm(), cfg, halo=build_halo(dg, Comm()))
    assert got.modularity == ref.modularity
    assert torch.equal(got.communities, ref.communities)
