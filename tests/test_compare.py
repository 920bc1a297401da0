"""Unit tests for the ground-truth comparison metrics (ref compare.cpp)."""

import pytest
import torch

from cuvite_amd.compare import compare_communities, gini_coefficient


def test_identical_clusterings():
    t = torch.tensor([0, 0, 1, 1, 2, 2])
    m = compare_communities(t, t)
    assert m["f_score"] == pytest.approx(1.0)
    assert m["precision"] == pytest.approx(1.0)
    assert m["recall"] == pytest.approx(1.0)
    assert m["f_mean"] == pytest.approx(1.0)
    assert m["gini_truth"] == m["gini_pred"]


def test_label_permutation_invariant():
    t = torch.tensor([0, 0, 1, 1, 2, 2])
    p = torch.tensor([7, 7, 3, 3, 9, 9])  # same partition, different labels
    m = compare_communities(t, p)
    assert m["f_score"] == pytest.approx(1.0)


def test_merged_communities_recall_vs_precision():
    # prediction merges the two truth communities: recall perfect per truth
    # community, precision halves
    t = torch.tensor([0, 0, 1, 1])
    p = torch.tensor([5, 5, 5, 5])
    m = compare_communities(t, p)
    assert m["recall"] == pytest.approx(1.0)
    assert m["precision"] == pytest.approx(0.5)
    assert m["f_score"] == pytest.approx(2 * 0.5 * 1.0 / 1.5)


def test_split_communities():
    # prediction splits one truth community in two
    t = torch.tensor([0, 0, 0, 0])
    p = torch.tensor([1, 1, 2, 2])
    m = compare_communities(t, p)
    assert m["recall"] == pytest.approx(0.5)
    assert m["precision"] == pytest.approx(1.0)


def test_gini():
    import numpy as np
    # all communities equal size -> 0
    assert gini_coefficient(np.array([0, 0, 1, 1, 2, 2])) == pytest.approx(0.0)
    # extreme skew -> approaches (n-1)/n normalization
    skew = np.array([0] * 99 + [1])
    assert gini_coefficient(skew) > 0.4
