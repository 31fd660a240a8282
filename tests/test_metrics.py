"""Metric parity tests vs hand-computed values and (when available) sklearn
(reference main.py:300-359)."""

import numpy as np
import pytest

from code2vec_amd.data.vocab import Vocab
from code2vec_amd.engine import metrics as M


@pytest.fixture
def vocab():
    v = Vocab()
    v.append("getvalue", subtokens=["get", "value"])
    v.append("setvalue", subtokens=["set", "value"])
    v.append("tostring", subtokens=["to", "string"])
    v.append("get", subtokens=["get"])
    return v


def test_subtoken_match_hand_computed(vocab):
    # expected: getvalue(2 toks), tostring(2); actual: setvalue, tostring
    exp = [0, 2]
    act = [1, 2]
    # matches: "value" in setvalue (1), to+string (2) => m=3, E=4, A=4
    acc, prec, rec, f1 = M.subtoken_match(exp, act, vocab)
    assert acc == pytest.approx(3 / (4 + 4 - 3))
    assert prec == pytest.approx(3 / 4)
    assert rec == pytest.approx(3 / 4)
    assert f1 == pytest.approx(3 / 4)


def test_subtoken_counts_compose(vocab):
    exp = [0, 2, 1, 3]
    act = [1, 2, 1, 0]
    whole = M.subtoken_match(exp, act, vocab)
    m1 = M.subtoken_match_counts(exp[:2], act[:2], vocab)
    m2 = M.subtoken_match_counts(exp[2:], act[2:], vocab)
    combined = M.subtoken_stats_from_counts(
        m1[0] + m2[0], m1[1] + m2[1], m1[2] + m2[2]
    )
    assert whole == pytest.approx(combined)


def test_averaged_subtoken_match(vocab):
    exp = [0]
    act = [1]
    acc, prec, rec, f1 = M.averaged_subtoken_match(exp, act, vocab)
    # match=1 ("value"); acc=1/(2+2-1), prec=rec=1/2, f1=1/2
    assert acc == pytest.approx(1 / 3)
    assert prec == pytest.approx(0.5)
    assert rec == pytest.approx(0.5)
    assert f1 == pytest.approx(0.5)


def test_exact_match_against_sklearn():
    sklearn = pytest.importorskip("sklearn.metrics")
    rng = np.random.default_rng(0)
    y_true = rng.integers(0, 5, 200)
    y_pred = rng.integers(0, 5, 200)
    acc, prec, rec, f1 = M.exact_match(y_true, y_pred)
    p, r, f, _ = sklearn.precision_recall_fscore_support(
        y_true, y_pred, average="weighted", zero_division=0
    )
    assert acc == pytest.approx(sklearn.accuracy_score(y_true, y_pred))
    assert prec == pytest.approx(p)
    assert rec == pytest.approx(r)
    assert f1 == pytest.approx(f)


def test_exact_match_perfect():
    acc, prec, rec, f1 = M.exact_match([1, 2, 3], [1, 2, 3])
    assert (acc, prec, rec, f1) == (1.0, 1.0, 1.0, 1.0)


def test_metrics_empty_inputs_return_zeros():
    """Empty eval sets must yield 0-tuples (never NaN — metric lines are
    JSON; the reference's np.average would produce NaN here)."""
    from code2vec_amd.data.vocab import Vocab

    v = Vocab()
    v.append("getfoo", subtokens=["get", "foo"])
    assert M.exact_match([], []) == (0.0, 0.0, 0.0, 0.0)
    assert M.subtoken_match([], [], v) == (0.0, 0.0, 0.0, 0.0)
    assert M.averaged_subtoken_match([], [], v) == (0.0, 0.0, 0.0, 0.0)
