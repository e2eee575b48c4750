"""Evaluation metrics: weighted precision/recall/F1 + accuracy.

Same metric suite as the reference notebook (cell 3: sklearn
precision_score/recall_score/f1_score with average='weighted' + accuracy).
"""
from __future__ import annotations

from typing import Dict

import torch


def classification_metrics(y_true: torch.Tensor, y_pred: torch.Tensor,
                           n_classes: int) -> Dict[str, float]:
    y_true = y_true.long().cpu()
    y_pred = y_pred.long().cpu()
    conf = torch.zeros(n_classes, n_classes, dtype=torch.double)
    for t, p in zip(y_true.tolist(), y_pred.tolist()):
        conf[t, p] += 1
    support = conf.sum(1)
    tp = conf.diag()
    pred_pos = conf.sum(0)
    prec_c = torch.where(pred_pos > 0, tp / pred_pos.clamp(min=1), torch.zeros(n_classes, dtype=torch.double))
    rec_c = torch.where(support > 0, tp / support.clamp(min=1), torch.zeros(n_classes, dtype=torch.double))
    f1_c = torch.where(prec_c + rec_c > 0, 2 * prec_c * rec_c / (prec_c + rec_c).clamp(min=1e-12),
                       torch.zeros(n_classes, dtype=torch.double))
    w = support / support.sum().clamp(min=1)
    return {
        "accuracy": float(tp.sum() / support.sum().clamp(min=1)),
        "precision": float((prec_c * w).sum()),
        "recall": float((rec_c * w).sum()),
        "f1": float((f1_c * w).sum()),
    }
