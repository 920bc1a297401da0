"""CLI driver tests (ref main.cpp:63-585: the graphClustering binary)."""

import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_cli(args, cwd):
    env = dict(os.environ, PYTHONPATH=REPO)
    return subprocess.run(
        [sys.executable, "-m", "cuvite_amd", "--device", "cpu"] + args,
        capture_output=True, text=True, cwd=cwd, env=env, timeout=300)


def test_karate_plain(tmp_path):
    r = run_cli(["--karate"], tmp_path)
    assert r.returncode == 0, r.stderr
    assert "Final modularity" in r.stdout
    q = float([ln for ln in r.stdout.splitlines()
               if ln.startswith("Final modularity")][0].split()[-1])
    assert 0.35 <= q <= 0.45  # known karate-club Louvain range


def test_karate_output_and_ground_truth(tmp_path):
    # self-comparison: dump communities, re-use them as ground truth -> F=1
    r = run_cli(["--karate", "-o"], tmp_path)
    assert r.returncode == 0, r.stderr
    comm_file = tmp_path / "graph.communities"
    assert comm_file.exists()
    r2 = run_cli(["--karate", "-g", str(comm_file)], tmp_path)
    assert r2.returncode == 0, r2.stderr
    line = [ln for ln in r2.stdout.splitlines() if "f-score" in ln][0]
    f = float(line.split("f-score=")[1].split()[0])
    assert f == pytest.approx(1.0, abs=1e-9)


def test_ordering_variant(tmp_path):
    r = run_cli(["--karate", "-d", "4"], tmp_path)
    assert r.returncode == 0, r.stderr
    q = float([ln for ln in r.stdout.splitlines()
               if ln.startswith("Final modularity")][0].split()[-1])
    assert q >= 0.30


def test_coloring_variant(tmp_path):
    r = run_cli(["--karate", "-c", "4", "-i"], tmp_path)
    assert r.returncode == 0, r.stderr
    assert "Final modularity" in r.stdout


def test_early_term_types(tmp_path):
    for t, extra in (("1", []), ("2", ["-a", "0.5"]), ("3", []),
                     ("4", ["-a", "0.5"])):
        r = run_cli(["--karate", "-t", t] + extra, tmp_path)
        assert r.returncode == 0, (t, r.stderr)


def test_generate_write_reload(tmp_path):
    out = tmp_path / "gen.bin"
    r = run_cli(["-n", "512", "-e", "3", "-s", str(out), "-j"], tmp_path)
    assert r.returncode == 0, r.stderr
    assert out.exists()
    r2 = run_cli(["-f", str(out), "-p"], tmp_path)
    assert r2.returncode == 0, r2.stderr
    assert "Final modularity" in r2.stdout


def test_balanced_load(tmp_path):
    out = tmp_path / "gen.bin"
    r = run_cli(["-n", "512", "-s", str(out), "-j"], tmp_path)
    assert r.returncode == 0, r.stderr
    r2 = run_cli(["-f", str(out), "-b", "--stats"], tmp_path)
    assert r2.returncode == 0, r2.stderr
    assert "imbalance" in r2.stdout


def test_flag_validation(tmp_path):
    assert run_cli(["--karate", "-c", "2", "-d", "2"], tmp_path).returncode != 0
    assert run_cli(["--karate", "-p", "-i"], tmp_path).returncode != 0
    assert run_cli(["--karate", "-t", "7"], tmp_path).returncode != 0
    assert run_cli([], tmp_path).returncode != 0


def test_ordering_matches_plain_quality():
    """-d on one rank should reach comparable modularity to plain."""
    from cuvite_amd.generators import rmat_graph
    from cuvite_amd.graph import single_partition
    from cuvite_amd.louvain import louvain, LouvainConfig
    from cuvite_amd.parallel import Comm

    g = rmat_graph(9, 8, seed=3)
    dg = single_partition(g)
    qs = {}
    for name, cfg in (
            ("plain", LouvainConfig(backend="torch")),
            ("order", LouvainConfig(backend="torch", ordering=True,
                                    max_colors=6))):
        res = louvain(dg, Comm(torch.device("cpu")), cfg)
        qs[name] = res.modularity
    assert qs["order"] >= qs["plain"] - 0.05


def test_lfr_generator_recovery():
    """LFR planted communities are recovered at low mixing (acceptance-style
    test, ref README:105-117 ground-truth workflow)."""
    import torch
    from cuvite_amd.generators import lfr_graph
    from cuvite_amd.graph import single_partition
    from cuvite_amd.louvain import louvain, LouvainConfig
    from cuvite_amd.parallel import Comm
    from cuvite_amd.compare import compare_communities

    g, truth = lfr_graph(2000, mu=0.2, seed=5)
    res = louvain(single_partition(g), Comm(torch.device("cpu")),
                  LouvainConfig(backend="torch"))
    m = compare_communities(truth, res.communities)
    assert m["recall"] > 0.85
    assert m["f_score"] > 0.8


def test_lfr_dist_slices_cover_graph():
    import torch
    from cuvite_amd.generators import lfr_graph, lfr_dist_graph
    g, truth = lfr_graph(1000, mu=0.4, seed=9)
    ne = 0
    for r in range(4):
        dg, truth_r = lfr_dist_graph(1000, r, 4, mu=0.4, seed=9)
        assert torch.equal(truth_r, truth)
        ne += dg.ne
    assert ne == g.ne


def test_vertex_order_degree(tmp_path):
    """--vertex-order degree relabels internally but dumps communities in
    the INPUT vertex order and id space: self-ground-truth comparison of a
    natural-order dump against a degree-order run's clustering is label-
    invariant, and the dumped labels must be valid input-space gids."""
    r = run_cli(["--karate", "-o", "--vertex-order", "degree"], tmp_path)
    assert r.returncode == 0, r.stderr
    q = float([ln for ln in r.stdout.splitlines()
               if ln.startswith("Final modularity")][0].split()[-1])
    assert q >= 0.35
    comm_file = tmp_path / "graph.communities"
    labels = [int(ln.split()[-1]) for ln in
              comm_file.read_text().splitlines() if ln.strip()]
    assert len(labels) == 34
    assert all(0 <= c < 34 for c in labels)
    # degree-order clustering vs itself through the -g path -> F = 1
    r2 = run_cli(["--karate", "-g", str(comm_file),
                  "--vertex-order", "degree"], tmp_path)
    assert r2.returncode == 0, r2.stderr
    line = [ln for ln in r2.stdout.splitlines() if "f-score" in ln][0]
    f = float(line.split("f-score=")[1].split()[0])
    assert f > 0.9  # same algorithm, same order -> same clustering
