"""Ground-truth community comparison: all-pairs F-score and Gini coefficient.

Reference: compare_communities / compute_gini_coeff (compare.cpp:8-286).
The reference computes, for every (ground-truth community, computed community)
pair sharing vertices, precision = |intersection|/|computed|, recall =
|intersection|/|truth|, F = 2pr/(p+r), and reports the mean over truth
communities of the best F (plus the same in the other direction), and the
Gini coefficient of the community-size distribution.
"""

from __future__ import annotations

import numpy as np
import torch


def _best_f_mean(truth: np.ndarray, pred: np.ndarray):
    """Mean over truth communities of the best F-score against pred, plus
    the mean precision/recall of those best-F pairs."""
    # contingency via sparse counting on (truth, pred) pairs
    t_ids, t_inv = np.unique(truth, return_inverse=True)
    p_ids, p_inv = np.unique(pred, return_inverse=True)
    nt, npp = len(t_ids), len(p_ids)
    key = t_inv.astype(np.int64) * npp + p_inv
    uk, cnt = np.unique(key, return_counts=True)
    ti = uk // npp
    pi = uk % npp
    t_sizes = np.bincount(t_inv, minlength=nt)
    p_sizes = np.bincount(p_inv, minlength=npp)
    prec = cnt / p_sizes[pi]
    rec = cnt / t_sizes[ti]
    f = 2 * prec * rec / np.maximum(prec + rec, 1e-300)
    best = np.zeros(nt)
    np.maximum.at(best, ti, f)
    # precision/recall of the pair achieving the best F per truth community:
    # sort pairs by (ti, f) and take each ti-segment's last entry (the argmax;
    # explicit reduction — duplicate-fancy-index assignment order is
    # unspecified in NumPy). Every truth community appears in >=1 pair, so
    # the segment ends enumerate all nt communities in order.
    order = np.lexsort((f, ti))
    ti_s = ti[order]
    seg_end = np.flatnonzero(np.r_[ti_s[1:] != ti_s[:-1], True])
    best_pair = order[seg_end]
    return float(best.mean()), float(prec[best_pair].mean()), \
        float(rec[best_pair].mean())


def compare_communities(truth: torch.Tensor, pred: torch.Tensor) -> dict:
    """Returns {'f_truth_to_pred', 'f_pred_to_truth', 'f_mean',
    'gini_truth', 'gini_pred', 'n_truth', 'n_pred'}."""
    t = truth.cpu().numpy()
    p = pred.cpu().numpy()
    if t.shape != p.shape:  # ground-truth files may omit trailing vertices
        n = min(len(t), len(p))
        t, p = t[:n], p[:n]
    f_tp, prec, rec = _best_f_mean(t, p)
    f_pt, _, _ = _best_f_mean(p, t)
    return {
        "f_truth_to_pred": f_tp,
        "f_pred_to_truth": f_pt,
        "f_mean": 0.5 * (f_tp + f_pt),
        "precision": prec,
        "recall": rec,
        "f_score": f_tp,
        "gini_truth": gini_coefficient(t),
        "gini_pred": gini_coefficient(p),
        "n_truth": int(len(np.unique(t))),
        "n_pred": int(len(np.unique(p))),
    }


def gini_coefficient(labels: np.ndarray) -> float:
    """Gini coefficient of the community-size distribution
    (ref compute_gini_coeff, compare.cpp:260-286)."""
    _, sizes = np.unique(labels, return_counts=True)
    sizes = np.sort(sizes).astype(np.float64)
    n = len(sizes)
    if n == 0 or sizes.sum() == 0:
        return 0.0
    cum = np.cumsum(sizes)
    return float((n + 1 - 2 * (cum / cum[-1]).sum()) / n)
