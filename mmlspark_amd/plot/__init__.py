"""Matplotlib evaluation plots (analog of mmlspark/plot/plot.py:17,45).

Same two-call surface as the reference — ``confusionMatrix(df, y, yhat,
labels)`` and ``roc(df, y, yhat)`` — drawing onto the current matplotlib
axes so callers compose them into their own figures.  Metrics come from
our own ComputeModelStatistics math (no sklearn requirement at runtime,
though sklearn is used in tests as an oracle).
"""
from __future__ import annotations

import itertools

import numpy as np

__all__ = ["confusionMatrix", "roc"]


def _confusion_counts(y, y_hat, n):
    cm = np.zeros((n, n), dtype=np.int64)
    for t, p in zip(np.asarray(y, dtype=np.int64),
                    np.asarray(y_hat, dtype=np.int64)):
        cm[t, p] += 1
    return cm


def confusionMatrix(df, y_col: str, y_hat_col: str, labels):
    """Draw a row-normalised confusion matrix with per-cell counts and an
    accuracy banner (plot/plot.py:17-43)."""
    import matplotlib.pyplot as plt

    y = df[y_col].to_numpy()
    y_hat = df[y_hat_col].to_numpy()
    accuracy = float(np.mean(y == y_hat))
    cm = _confusion_counts(y, y_hat, len(labels))
    with np.errstate(invalid="ignore"):
        cmn = cm.astype(np.float64) / np.maximum(cm.sum(axis=1, keepdims=True), 1)
    plt.text(-.3, -.55, "$Accuracy$ $=$ ${}\\%$".format(round(accuracy * 100, 1)),
             fontsize=18)
    ticks = np.arange(len(labels))
    plt.xticks(ticks, labels, rotation=0)
    plt.yticks(ticks, labels, rotation=90)
    plt.imshow(cmn, interpolation="nearest", cmap=plt.cm.Blues, vmin=0, vmax=1)
    for i, j in itertools.product(range(cm.shape[0]), range(cm.shape[1])):
        plt.text(j, i, cm[i, j], horizontalalignment="center", fontsize=18,
                 color="white" if cmn[i, j] > .1 else "black")
    plt.colorbar()
    plt.xlabel("Predicted Label", fontsize=18)
    plt.ylabel("True Label", fontsize=18)
    return cm


def roc(df, y_col: str, y_hat_col: str, thresh: float = .5):
    """Plot the ROC curve of scores against binarised labels and return its
    AUC (plot/plot.py:45-63)."""
    import matplotlib.pyplot as plt

    y = (df[y_col].to_numpy() > thresh).astype(np.int64)
    scores = df[y_hat_col].to_numpy().astype(np.float64)
    order = np.argsort(-scores)
    y = y[order]
    tp = np.concatenate([[0], np.cumsum(y)])
    fp = np.concatenate([[0], np.cumsum(1 - y)])
    P = max(int(tp[-1]), 1)
    N = max(int(fp[-1]), 1)
    tpr, fpr = tp / P, fp / N
    auc = float(np.trapezoid(tpr, fpr))
    plt.plot(fpr, tpr)
    plt.xlabel("False Positive Rate", fontsize=20)
    plt.ylabel("True Positive Rate", fontsize=20)
    plt.title("ROC Curve (AUC = {:.3f})".format(auc), fontsize=20)
    return auc
