"""Evaluation metrics — exact / subtoken / averaged-subtoken match.

Semantics replicate reference main.py:300-359 exactly; subtoken-F1
(``--eval_method subtoken``, the default) is the north-star quality metric.
Implemented without sklearn (the exact-match weighted P/R/F1 is computed
directly with numpy, matching sklearn's 'weighted' average semantics,
including the "actual-class precision" convention).

All three return ``(accuracy, precision, recall, f1)``.
"""

from __future__ import annotations

from typing import Sequence

import numpy as np


def exact_match(expected_labels, actual_labels):
    """Weighted P/R/F1 + accuracy over class labels (reference main.py:300-305).

    Matches sklearn precision_recall_fscore_support(average='weighted') +
    accuracy_score: per-class P/R/F1 weighted by true-class support
    (classes present in expected only; zero-division -> 0).
    """
    y_true = np.asarray(expected_labels, dtype=np.int64)
    y_pred = np.asarray(actual_labels, dtype=np.int64)
    classes = np.unique(np.concatenate([y_true, y_pred]))
    precisions = []
    recalls = []
    f1s = []
    supports = []
    for c in classes:
        tp = int(np.sum((y_true == c) & (y_pred == c)))
        fp = int(np.sum((y_true != c) & (y_pred == c)))
        fn = int(np.sum((y_true == c) & (y_pred != c)))
        support = tp + fn
        p = tp / (tp + fp) if (tp + fp) > 0 else 0.0
        r = tp / (tp + fn) if (tp + fn) > 0 else 0.0
        f = 2 * p * r / (p + r) if (p + r) > 0 else 0.0
        precisions.append(p)
        recalls.append(r)
        f1s.append(f)
        supports.append(support)
    supports = np.asarray(supports, dtype=np.float64)
    total = supports.sum()
    if total == 0:
        return 0.0, 0.0, 0.0, 0.0
    w = supports / total
    precision = float(np.sum(w * np.asarray(precisions)))
    recall = float(np.sum(w * np.asarray(recalls)))
    f1 = float(np.sum(w * np.asarray(f1s)))
    accuracy = float(np.mean(y_true == y_pred))
    return accuracy, precision, recall, f1


def subtoken_match(expected_labels, actual_labels, label_vocab):
    """Micro subtoken overlap (reference main.py:339-359):
    acc = m/(E+A-m), prec = m/A, rec = m/E, F1 harmonic."""
    match = 0.0
    expected_count = 0.0
    actual_count = 0.0
    itosub = label_vocab.itosubtokens
    for expected, actual in zip(_tolist(expected_labels), _tolist(actual_labels)):
        exp_subtokens = itosub[int(expected)]
        act_subtokens = itosub[int(actual)]
        for subtoken in exp_subtokens:
            if subtoken in act_subtokens:
                match += 1
        expected_count += len(exp_subtokens)
        actual_count += len(act_subtokens)
    return _micro_stats(match, expected_count, actual_count)


def subtoken_match_counts(expected_labels, actual_labels, label_vocab):
    """(match, expected_count, actual_count) — the all-reducible form used
    by DP evaluation; combine shards then call
    :func:`subtoken_stats_from_counts`."""
    match = 0.0
    expected_count = 0.0
    actual_count = 0.0
    itosub = label_vocab.itosubtokens
    for expected, actual in zip(_tolist(expected_labels), _tolist(actual_labels)):
        exp_subtokens = itosub[int(expected)]
        act_subtokens = itosub[int(actual)]
        for subtoken in exp_subtokens:
            if subtoken in act_subtokens:
                match += 1
        expected_count += len(exp_subtokens)
        actual_count += len(act_subtokens)
    return match, expected_count, actual_count


def subtoken_stats_from_counts(match, expected_count, actual_count):
    return _micro_stats(match, expected_count, actual_count)


def _micro_stats(match, expected_count, actual_count):
    denom = expected_count + actual_count - match
    accuracy = match / denom if denom > 0 else 0.0
    precision = match / actual_count if actual_count > 0 else 0.0
    recall = match / expected_count if expected_count > 0 else 0.0
    if precision + recall > 0:
        f1 = 2.0 * precision * recall / (precision + recall)
    else:
        f1 = 0.0
    return accuracy, precision, recall, f1


def averaged_subtoken_match(expected_labels, actual_labels, label_vocab):
    """Per-sample subtoken stats, averaged (reference main.py:308-336)."""
    accs, precs, recs, f1s = [], [], [], []
    itosub = label_vocab.itosubtokens
    for expected, actual in zip(_tolist(expected_labels), _tolist(actual_labels)):
        exp_subtokens = itosub[int(expected)]
        act_subtokens = itosub[int(actual)]
        match = 0
        for subtoken in exp_subtokens:
            if subtoken in act_subtokens:
                match += 1
        acc = match / float(len(exp_subtokens) + len(act_subtokens) - match)
        rec = match / float(len(exp_subtokens))
        prec = match / float(len(act_subtokens))
        f1 = 2.0 * prec * rec / (prec + rec) if (prec + rec) > 0 else 0.0
        accs.append(acc)
        precs.append(prec)
        recs.append(rec)
        f1s.append(f1)
    if not accs:
        # empty eval set: the reference would emit NaN (np.average of
        # empty), which is not valid JSON in the metric lines — report 0s
        return (0.0, 0.0, 0.0, 0.0)
    return (
        float(np.average(accs)),
        float(np.average(precs)),
        float(np.average(recs)),
        float(np.average(f1s)),
    )


def _tolist(x) -> Sequence:
    if hasattr(x, "tolist"):
        return x.tolist()
    return list(x)
