"""GLUE fine-tuning metrics + scorer.

Capability parity with the reference's GLUE tooling
(/root/reference/BERT/bert/compute_glue_scores.py and the sources.py/text.py
data utils): per-task metrics (accuracy, F1, Matthews corrcoef,
Pearson/Spearman), TSV prediction/label loading, and a task->metric map.
Dataset download is out of scope offline; the scorer operates on local TSVs
or in-memory arrays.
"""
from __future__ import annotations

import csv
import math
from typing import Dict, List, Sequence

import numpy as np

TASK_METRICS = {
    "cola": ("mcc",),
    "sst-2": ("acc",),
    "mrpc": ("acc", "f1"),
    "sts-b": ("pearson", "spearman"),
    "qqp": ("acc", "f1"),
    "mnli": ("acc",),
    "qnli": ("acc",),
    "rte": ("acc",),
    "wnli": ("acc",),
}


def accuracy(preds: Sequence, labels: Sequence) -> float:
    p = np.asarray(preds)
    l = np.asarray(labels)
    return float((p == l).mean())


def f1(preds: Sequence, labels: Sequence, positive=1) -> float:
    p = np.asarray(preds) == positive
    l = np.asarray(labels) == positive
    tp = float((p & l).sum())
    fp = float((p & ~l).sum())
    fn = float((~p & l).sum())
    if tp == 0:
        return 0.0
    prec = tp / (tp + fp)
    rec = tp / (tp + fn)
    return 2 * prec * rec / (prec + rec)


def matthews_corrcoef(preds: Sequence, labels: Sequence) -> float:
    p = np.asarray(preds).astype(int)
    l = np.asarray(labels).astype(int)
    tp = float(((p == 1) & (l == 1)).sum())
    tn = float(((p == 0) & (l == 0)).sum())
    fp = float(((p == 1) & (l == 0)).sum())
    fn = float(((p == 0) & (l == 1)).sum())
    denom = math.sqrt((tp + fp) * (tp + fn) * (tn + fp) * (tn + fn))
    if denom == 0:
        return 0.0
    return (tp * tn - fp * fn) / denom


def pearson(preds: Sequence, labels: Sequence) -> float:
    from scipy import stats

    return float(stats.pearsonr(np.asarray(preds, float), np.asarray(labels, float))[0])


def spearman(preds: Sequence, labels: Sequence) -> float:
    from scipy import stats

    return float(stats.spearmanr(np.asarray(preds, float), np.asarray(labels, float))[0])


_METRIC_FNS = {
    "acc": accuracy,
    "f1": f1,
    "mcc": matthews_corrcoef,
    "pearson": pearson,
    "spearman": spearman,
}


def compute_glue_scores(task: str, preds: Sequence, labels: Sequence) -> Dict[str, float]:
    task = task.lower()
    if task not in TASK_METRICS:
        raise ValueError(f"unknown GLUE task {task!r}")
    return {m: _METRIC_FNS[m](preds, labels) for m in TASK_METRICS[task]}


def load_tsv_column(path: str, column, has_header: bool = True) -> List[str]:
    """Load one column (index or name) from a GLUE TSV file."""
    out: List[str] = []
    with open(path, newline="", encoding="utf-8") as f:
        reader = csv.reader(f, delimiter="\t", quotechar=None)
        header = next(reader) if has_header else None
        idx = column if isinstance(column, int) else header.index(column)
        for row in reader:
            if len(row) > idx:
                out.append(row[idx])
    return out


# -- per-task example loading (reference's DataProcessor subclasses,
# compute_glue_scores.py:199-524) — one table-driven loader instead of nine
# near-identical classes.  Columns follow the public GLUE TSV layouts:
# (text_a_col, text_b_col|None, label_col, labels|None for regression,
# has_header, test text_a/text_b cols — test TSVs lead with an id column).
_TASK_SPECS = {
    "mrpc":  {"a": 3, "b": 4,    "label": 0,  "labels": ["0", "1"],
              "header": True,  "test_a": 3, "test_b": 4},
    "mnli":  {"a": 8, "b": 9,    "label": -1,
              "labels": ["contradiction", "entailment", "neutral"],
              "header": True,  "test_a": 8, "test_b": 9},
    "cola":  {"a": 3, "b": None, "label": 1,  "labels": ["0", "1"],
              "header": False, "test_a": 1, "test_b": None},
    "sst-2": {"a": 0, "b": None, "label": 1,  "labels": ["0", "1"],
              "header": True,  "test_a": 1, "test_b": None},
    "sts-b": {"a": 7, "b": 8,    "label": -1, "labels": None,
              "header": True,  "test_a": 7, "test_b": 8},
    "qqp":   {"a": 3, "b": 4,    "label": 5,  "labels": ["0", "1"],
              "header": True,  "test_a": 1, "test_b": 2},
    "qnli":  {"a": 1, "b": 2,    "label": -1,
              "labels": ["entailment", "not_entailment"],
              "header": True,  "test_a": 1, "test_b": 2},
    "rte":   {"a": 1, "b": 2,    "label": -1,
              "labels": ["entailment", "not_entailment"],
              "header": True,  "test_a": 1, "test_b": 2},
    "wnli":  {"a": 1, "b": 2,    "label": -1, "labels": ["0", "1"],
              "header": True,  "test_a": 1, "test_b": 2},
}


def task_labels(task: str):
    """Label vocabulary for a classification task (None for STS-B
    regression) — reference get_labels()."""
    return _TASK_SPECS[task.lower()]["labels"]


def load_examples(task: str, tsv_path: str, set_type: str = "dev"
                  ) -> List[Dict[str, object]]:
    """Read one GLUE split TSV into [{text_a, text_b, label}] dicts
    (reference _create_examples of each DataProcessor).  `label` is the
    label-vocabulary index (classification), a float (sts-b), or None for
    test splits (unlabeled)."""
    spec = _TASK_SPECS[task.lower()]
    test = set_type == "test"
    a_col = spec["test_a"] if test else spec["a"]
    b_col = spec["test_b"] if test else spec["b"]
    out: List[Dict[str, object]] = []
    with open(tsv_path, newline="", encoding="utf-8") as f:
        reader = csv.reader(f, delimiter="\t", quotechar=None)
        if spec["header"] or test:  # test splits always carry a header row
            next(reader, None)
        for line in reader:
            # guard every column we will index, including the label column —
            # real QQP train.tsv contains short malformed rows (the reference
            # QqpProcessor wraps row access in try/except IndexError)
            need = max(a_col, b_col or 0)
            if not test:
                need = max(need, spec["label"])
            if len(line) <= need:
                continue
            label = None
            if not test:
                raw = line[spec["label"]]
                if spec["labels"] is None:
                    label = float(raw)
                else:
                    label = spec["labels"].index(raw)
            out.append({
                "text_a": line[a_col],
                "text_b": line[b_col] if b_col is not None else None,
                "label": label,
            })
    return out


def score_files(task: str, pred_file: str, label_file: str,
                pred_col=-1, label_col=-1) -> Dict[str, float]:
    preds = load_tsv_column(pred_file, pred_col)
    labels = load_tsv_column(label_file, label_col)
    if task.lower() == "sts-b":
        return compute_glue_scores(task, [float(x) for x in preds],
                                   [float(x) for x in labels])
    uniq = sorted(set(labels) | set(preds))
    to_id = {v: i for i, v in enumerate(uniq)}
    return compute_glue_scores(
        task, [to_id[x] for x in preds], [to_id[x] for x in labels]
    )
