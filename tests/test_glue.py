import numpy as np
import pytest

from oktopk_amd.glue import (accuracy, compute_glue_scores, f1,
                             matthews_corrcoef, pearson, spearman)


def test_accuracy_f1():
    assert accuracy([1, 0, 1], [1, 1, 1]) == pytest.approx(2 / 3)
    assert f1([1, 1, 0, 0], [1, 0, 1, 0]) == pytest.approx(0.5)


def test_mcc_perfect_and_random():
    assert matthews_corrcoef([1, 0, 1, 0], [1, 0, 1, 0]) == 1.0
    assert matthews_corrcoef([1, 1, 0, 0], [1, 0, 1, 0]) == 0.0


def test_correlations():
    x = [1.0, 2.0, 3.0, 4.0]
    assert pearson(x, [2.0, 4.0, 6.0, 8.0]) == pytest.approx(1.0)
    assert spearman(x, [1.0, 3.0, 9.0, 27.0]) == pytest.approx(1.0)


def test_task_map():
    s = compute_glue_scores("mrpc", [1, 0, 1], [1, 1, 1])
    assert set(s) == {"acc", "f1"}
    with pytest.raises(ValueError):
        compute_glue_scores("nope", [], [])


def test_metric_edge_cases():
    """Degenerate inputs: constant predictions (MCC denominator 0), perfect
    F1, anti-correlated pearson."""
    from oktopk_amd.glue import matthews_corrcoef, f1, pearson

    assert matthews_corrcoef([1, 1, 1, 1], [0, 1, 0, 1]) == 0.0  # undefined -> 0
    assert f1([1, 0, 1], [1, 0, 1]) == 1.0
    assert f1([0, 0, 0], [1, 1, 1]) == 0.0
    assert abs(pearson([1, 2, 3], [3, 2, 1]) + 1.0) < 1e-9
