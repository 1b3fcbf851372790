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


def test_load_examples_per_task(tmp_path):
    """Table-driven GLUE TSV loaders (reference DataProcessor subclasses,
    compute_glue_scores.py:199-524): column layout + label vocabulary."""
    from oktopk_amd.glue import load_examples, task_labels

    # MRPC dev: label, id1, id2, s1, s2 (header row)
    mrpc = tmp_path / "mrpc_dev.tsv"
    mrpc.write_text("Quality\t#1 ID\t#2 ID\t#1 String\t#2 String\n"
                    "1\ta\tb\tfirst sentence\tsecond sentence\n"
                    "0\tc\td\tthird\tfourth\n")
    ex = load_examples("mrpc", str(mrpc))
    assert len(ex) == 2
    assert ex[0] == {"text_a": "first sentence", "text_b": "second sentence",
                     "label": 1}
    assert ex[1]["label"] == 0

    # CoLA dev: no header, cols = source, label, star, sentence
    cola = tmp_path / "cola_dev.tsv"
    cola.write_text("src\t1\t*\tthe cat sat\nsrc\t0\t*\tcat the sat the\n")
    ex = load_examples("cola", str(cola))
    assert [e["label"] for e in ex] == [1, 0]
    assert ex[0]["text_a"] == "the cat sat"
    assert ex[0]["text_b"] is None

    # STS-B: regression label in the last column
    sts = tmp_path / "sts_dev.tsv"
    sts.write_text("index\tgenre\tfile\tyear\told1\told2\tsrc\ts1\ts2\tscore\n"
                   "0\tg\tf\ty\t-\t-\t-\ta sent\tb sent\t3.8\n")
    ex = load_examples("sts-b", str(sts))
    assert ex[0]["label"] == 3.8 and ex[0]["text_b"] == "b sent"
    assert task_labels("sts-b") is None

    # MNLI dev: textual labels -> vocabulary index
    mnli = tmp_path / "mnli_dev.tsv"
    header = "\t".join(f"c{i}" for i in range(10)) + "\tgold_label\n"
    mnli.write_text(header +
                    "\t".join("x" * 1 for _ in range(8)) +
                    "\tpremise\thypothesis\tentailment\n")
    ex = load_examples("mnli", str(mnli))
    assert ex[0]["label"] == task_labels("mnli").index("entailment")

    # test split: unlabeled, id-leading columns
    qnli_t = tmp_path / "qnli_test.tsv"
    qnli_t.write_text("index\tquestion\tsentence\n"
                      "0\twho?\tthe answer.\n")
    ex = load_examples("qnli", str(qnli_t), set_type="test")
    assert ex[0] == {"text_a": "who?", "text_b": "the answer.", "label": None}


def test_load_examples_malformed_rows(tmp_path):
    """Short rows are skipped, not crashed on — real QQP train.tsv contains
    rows shorter than the label column (reference QqpProcessor wraps row
    access in try/except IndexError); empty files must not StopIteration."""
    from oktopk_amd.glue import load_examples

    # QQP layout: id, qid1, qid2, question1, question2, is_duplicate(=col 5).
    # Row 2 has the text columns but NOT the label column.
    qqp = tmp_path / "qqp_train.tsv"
    qqp.write_text("id\tqid1\tqid2\tquestion1\tquestion2\tis_duplicate\n"
                   "0\ta\tb\tq one\tq two\t1\n"
                   "1\tc\td\tshort row q1\tshort row q2\n")
    ex = load_examples("qqp", str(qqp), set_type="train")
    assert len(ex) == 1 and ex[0]["label"] == 1

    empty = tmp_path / "empty.tsv"
    empty.write_text("")
    assert load_examples("qqp", str(empty), set_type="train") == []
