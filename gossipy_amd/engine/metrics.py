"""Batched evaluation metrics (the K13 site of SURVEY.md §2.4).

The reference calls sklearn once per node per round
(gossipy/model/handler.py:282-334). The engine evaluates every sampled
node's model in one batched pass: scores come from the backend
(``[R, n, k]`` class scores or ``[R, n, 1]`` margins), metrics are computed
with tensor ops on the same device, and only the tiny per-node metric dict
list returns to the host. In multi-GPU runs each rank evaluates its
resident sample and rank 0 gathers the dicts.
"""

from __future__ import annotations

from typing import Dict, List

import torch

__all__ = ["classification_metrics_shared", "binary_margin_metrics"]


def _macro_prf(conf: torch.Tensor):
    """Macro precision/recall/F1 from per-node confusion ``[R, k, k]``
    (true x pred), with sklearn's zero_division=0 convention."""
    tp = torch.diagonal(conf, dim1=1, dim2=2)  # [R, k]
    pred_tot = conf.sum(dim=1)  # [R, k] column sums
    true_tot = conf.sum(dim=2)  # [R, k] row sums
    prec = torch.where(pred_tot > 0, tp / pred_tot.clamp(min=1), torch.zeros_like(tp))
    rec = torch.where(true_tot > 0, tp / true_tot.clamp(min=1), torch.zeros_like(tp))
    denom = prec + rec
    f1 = torch.where(denom > 0, 2 * prec * rec / denom.clamp(min=1e-12), torch.zeros_like(tp))
    return prec.mean(dim=1), rec.mean(dim=1), f1.mean(dim=1)


def _rank_auc(scores: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
    """Per-node ROC-AUC of positive-class scores ``[R, n]`` against shared
    binary labels ``[n]`` via the Mann-Whitney rank statistic (average ranks
    for ties)."""
    R, n = scores.shape
    dev = scores.device
    order = scores.argsort(dim=1)
    ss = scores.gather(1, order)
    # average tied ranks, fully vectorized (no host sync, exact): group
    # equal sorted scores, segment-mean their 1-based positions
    new_grp = torch.ones(R, n, dtype=torch.bool, device=dev)
    new_grp[:, 1:] = ss[:, 1:] != ss[:, :-1]
    gid = new_grp.to(torch.int64).cumsum(dim=1) - 1  # [R, n]
    pos = (
        torch.arange(1, n + 1, dtype=torch.float64, device=dev)
        .expand(R, n)
    )
    gsum = torch.zeros(R, n, dtype=torch.float64, device=dev)
    gsum.scatter_add_(1, gid, pos)
    gcnt = torch.zeros(R, n, dtype=torch.float64, device=dev)
    gcnt.scatter_add_(1, gid, torch.ones_like(pos))
    avg = (gsum / gcnt.clamp(min=1)).gather(1, gid)  # sorted-order ranks
    ranks = torch.empty(R, n, dtype=scores.dtype, device=dev)
    ranks.scatter_(1, order, avg.to(scores.dtype))
    pos = y > 0.5 if y.max() <= 1 else y > 0
    npos = int(pos.sum())
    nneg = n - npos
    if npos == 0 or nneg == 0:
        return torch.full((R,), 0.5, device=scores.device)
    rank_sum = ranks[:, pos].sum(dim=1)
    return (rank_sum - npos * (npos + 1) / 2) / (npos * nneg)


def classification_metrics_shared(
    scores: torch.Tensor, y: torch.Tensor, with_auc: bool = True
) -> List[Dict[str, float]]:
    """Metrics of R node models on one shared eval set.

    ``scores``: ``[R, n, k]`` class scores; ``y``: ``[n]`` class indices.
    Returns one dict per node (accuracy, macro precision/recall/F1, AUC for
    binary k=2 — matching gossipy/model/handler.py:282-334).
    """
    R, n, k = scores.shape
    y = y.long()
    pred = scores.argmax(dim=2)  # [R, n]
    acc = (pred == y.unsqueeze(0)).float().mean(dim=1)
    oh_pred = torch.nn.functional.one_hot(pred, k).to(scores.dtype)  # [R,n,k]
    oh_true = torch.nn.functional.one_hot(y, k).to(scores.dtype)  # [n,k]
    conf = torch.einsum("nt,rnp->rtp", oh_true, oh_pred)  # [R, k, k]
    prec, rec, f1 = _macro_prf(conf)
    out = []
    auc = None
    if with_auc and k == 2:
        classes = torch.unique(y)
        if len(classes) == 2:
            auc = _rank_auc(scores[:, :, 1], y.to(scores.dtype))
        else:
            auc = torch.full((R,), 0.5, device=scores.device)
    accs, precs, recs, f1s = acc.tolist(), prec.tolist(), rec.tolist(), f1.tolist()
    aucs = auc.tolist() if auc is not None else None
    for r in range(R):
        d = {
            "accuracy": accs[r],
            "precision": precs[r],
            "recall": recs[r],
            "f1_score": f1s[r],
        }
        if aucs is not None:
            d["auc"] = aucs[r]
        out.append(d)
    return out


def binary_margin_metrics(
    margins: torch.Tensor, y: torch.Tensor
) -> List[Dict[str, float]]:
    """Metrics for margin models (AdaLine/Pegasos): predictions are
    ``sign(margin)`` in {-1,+1}, labels are ±1
    (gossipy/model/handler.py:375-391)."""
    R, n = margins.shape[:2]
    m = margins.reshape(R, n)
    y = y.reshape(n)
    pred = torch.where(m >= 0, 1.0, -1.0)
    acc = (pred == y.unsqueeze(0)).float().mean(dim=1)
    # 2-class macro PRF over classes {-1, +1}
    conf = torch.zeros(R, 2, 2, device=m.device)
    yt = (y > 0).long()  # 0 = class -1, 1 = class +1
    pt = (pred > 0).long()
    oh_true = torch.nn.functional.one_hot(yt, 2).float()
    oh_pred = torch.nn.functional.one_hot(pt, 2).float()
    conf = torch.einsum("nt,rnp->rtp", oh_true, oh_pred)
    prec, rec, f1 = _macro_prf(conf)
    auc = _rank_auc(m, (y > 0).to(m.dtype))
    out = []
    for r in range(R):
        out.append(
            {
                "accuracy": float(acc[r]),
                "precision": float(prec[r]),
                "recall": float(rec[r]),
                "f1_score": float(f1[r]),
                "auc": float(auc[r]),
            }
        )
    return out
