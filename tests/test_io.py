import os

import pytest
import torch

from cuvite_amd.generators import karate_graph, rmat_graph
from cuvite_amd.io import (load_dist_graph, load_graph, write_graph,
                           read_header, load_ground_truth, write_communities)


def test_roundtrip_whole(tmp_path):
    g = karate_graph()
    p = str(tmp_path / "karate.bin")
    write_graph(p, g)
    assert read_header(p) == (34, 156)
    g2 = load_graph(p)
    assert torch.equal(g.rowptr, g2.rowptr)
    assert torch.equal(g.tails, g2.tails)
    assert torch.allclose(g.weights, g2.weights)


def test_sharded_load_matches_whole(tmp_path):
    g = rmat_graph(7, 8, seed=5)
    p = str(tmp_path / "rmat.bin")
    write_graph(p, g)
    for nranks in (2, 3, 4):
        off = 0
        for r in range(nranks):
            dg = load_dist_graph(p, r, nranks)
            nvl = dg.nv
            assert torch.equal(dg.g.rowptr + int(g.rowptr[off]),
                               g.rowptr[off:off + nvl + 1])
            e0, e1 = int(g.rowptr[off]), int(g.rowptr[off + nvl])
            assert torch.equal(dg.g.tails, g.tails[e0:e1])
            off += nvl
        assert off == g.nv


def test_balanced_load(tmp_path):
    g = rmat_graph(7, 8, seed=5)
    p = str(tmp_path / "rmat.bin")
    write_graph(p, g)
    shards = [load_dist_graph(p, r, 4, balanced=True) for r in range(4)]
    assert sum(s.nv for s in shards) == g.nv
    assert sum(s.ne for s in shards) == g.ne
    ne_max = max(s.ne for s in shards)
    # balanced shards should be closer to ne/4 than the worst-case skew
    assert ne_max <= g.ne  # sanity
    tails = torch.cat([s.g.tails for s in shards])
    assert torch.equal(tails, g.tails)


def test_unit_weights(tmp_path):
    g = karate_graph()
    p = str(tmp_path / "k.bin")
    write_graph(p, g)
    dg = load_dist_graph(p, 0, 1, unit_weights=True)
    assert float(dg.g.weights.sum()) == g.ne


def test_ground_truth_io(tmp_path):
    p = str(tmp_path / "gt.txt")
    comm = torch.tensor([0, 0, 1, 1, 2], dtype=torch.int64)
    write_communities(p, comm)
    gt = load_ground_truth(p)
    assert torch.equal(gt, comm)
    # 1-based variant
    with open(p, "w") as f:
        for v in range(5):
            f.write(f"{v+1} {int(comm[v])+1}\n")
    gt1 = load_ground_truth(p, zero_based=False)
    assert torch.equal(gt1, comm)


def test_empty_and_tiny_graphs(tmp_path):
    """Degenerate shapes: single vertex, no edges."""
    import numpy as np
    from cuvite_amd.graph import Graph, single_partition
    from cuvite_amd.louvain import louvain, LouvainConfig
    from cuvite_amd.parallel import Comm

    g = Graph(torch.tensor([0, 0], dtype=torch.int64),
              torch.zeros(0, dtype=torch.int64),
              torch.zeros(0, dtype=torch.float64))
    # isolated single vertex: modularity 0, stays singleton
    res = louvain(single_partition(g), Comm(torch.device("cpu")),
                  LouvainConfig(backend="torch", one_phase=True))
    assert res.communities.tolist() == [0]

    p = tmp_path / "tiny.bin"
    write_graph(str(p), g)
    g2 = load_graph(str(p))
    assert g2.nv == 1 and g2.ne == 0


def test_load_missing_file_raises(tmp_path):
    with pytest.raises(FileNotFoundError):
        load_graph(str(tmp_path / "nope.bin"))


def test_convert_edge_list(tmp_path):
    """Text edge list -> Vite binary -> Louvain end to end."""
    from cuvite_amd.convert import convert_edge_list
    from cuvite_amd.generators import karate_graph

    # write karate as a 1-based weighted edge list (each undirected edge once)
    g = karate_graph()
    lines = ["# karate club"]
    seen = set()
    for u in range(g.nv):
        for k in range(int(g.rowptr[u]), int(g.rowptr[u + 1])):
            v = int(g.tails[k])
            if (v, u) in seen or (u, v) in seen:
                continue
            seen.add((u, v))
            lines.append(f"{u + 1} {v + 1} {float(g.weights[k])}")
    src = tmp_path / "karate.txt"
    src.write_text("\n".join(lines) + "\n")
    out = tmp_path / "karate.bin"
    g2 = convert_edge_list(str(src), str(out), one_based=True)
    assert g2.nv == g.nv and g2.ne == g.ne
    g3 = load_graph(str(out))
    assert torch.equal(g3.rowptr, g.rowptr)

    from cuvite_amd.graph import single_partition
    from cuvite_amd.louvain import louvain, LouvainConfig
    from cuvite_amd.parallel import Comm
    res = louvain(single_partition(g3), Comm(torch.device("cpu")),
                  LouvainConfig(backend="torch"))
    assert res.modularity == pytest.approx(0.408695, abs=1e-6)
