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
