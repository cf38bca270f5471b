"""Fusion math parity with the reference semantics
(reference src/core/retrievers/hybrid.py:204-259)."""

import math

from sentio_amd.index.fusion import add_plugin_scores, fuse


DENSE = [("a", 0.9), ("b", 0.8), ("c", 0.5)]
SPARSE = [("b", 12.0), ("d", 7.0)]


def test_rrf_scores_exact():
    out = dict(fuse(DENSE, SPARSE, method="rrf", top_k=10, rrf_k=60))
    assert math.isclose(out["a"], 1 / 60)
    assert math.isclose(out["b"], 1 / 61 + 1 / 60)
    assert math.isclose(out["c"], 1 / 62)
    assert math.isclose(out["d"], 1 / 61)


def test_weighted_rrf_uses_weights():
    out = dict(fuse(DENSE, SPARSE, method="weighted_rrf", top_k=10, rrf_k=60,
                    dense_weight=0.7, sparse_weight=0.3))
    assert math.isclose(out["a"], 0.7 / 60)
    assert math.isclose(out["d"], 0.3 / 61)
    assert math.isclose(out["b"], 0.7 / 61 + 0.3 / 60)


def test_comb_sum_minmax_normalizes():
    out = dict(fuse(DENSE, SPARSE, method="comb_sum", top_k=10,
                    dense_weight=0.7, sparse_weight=0.3))
    # dense: a=1.0, b=0.75, c=0.0 after min-max; sparse: b=1.0, d=0.0
    assert math.isclose(out["a"], 0.7 * 1.0)
    assert math.isclose(out["b"], 0.7 * 0.75 + 0.3 * 1.0)
    assert math.isclose(out["c"], 0.0)
    assert math.isclose(out["d"], 0.0)


def test_comb_sum_all_equal_treated_as_one():
    # all-equal scores normalize to 1.0 (reference hybrid.py:216-218)
    out = dict(fuse([("a", 0.5), ("b", 0.5)], [], method="comb_sum",
                    top_k=10, dense_weight=1.0))
    assert out["a"] == out["b"] == 1.0


def test_rank_order_and_topk_truncation():
    out = fuse(DENSE, SPARSE, method="rrf", top_k=2, rrf_k=60)
    assert len(out) == 2
    assert out[0][0] == "b"  # appears in both lists


def test_unknown_method_raises():
    import pytest

    with pytest.raises(ValueError):
        fuse(DENSE, SPARSE, method="nope")


def test_plugin_scores_add_directly():
    fused = fuse(DENSE, SPARSE, method="rrf", top_k=10)
    out = dict(add_plugin_scores(fused, {"c": 5.0}))
    assert out["c"] > out["a"]
