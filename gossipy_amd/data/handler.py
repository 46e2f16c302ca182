"""Concrete data handlers (parity: gossipy/data/handler.py)."""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple, Union

import numpy as np
import torch

from . import DataHandler

__all__ = [
    "ClassificationDataHandler",
    "ClusteringDataHandler",
    "RegressionDataHandler",
    "RecSysDataHandler",
]


class ClassificationDataHandler(DataHandler):
    """Holds train/eval matrices with a seeded split
    (gossipy/data/handler.py:25-134)."""

    def __init__(
        self,
        X: Union[np.ndarray, torch.Tensor],
        y: Union[np.ndarray, torch.Tensor],
        X_te: Optional[Union[np.ndarray, torch.Tensor]] = None,
        y_te: Optional[Union[np.ndarray, torch.Tensor]] = None,
        test_size: float = 0.2,
        seed: int = 42,
    ):
        assert 0 <= test_size < 1
        assert isinstance(X, (torch.Tensor, np.ndarray))

        if test_size > 0 and (X_te is None or y_te is None):
            if isinstance(X, torch.Tensor):
                n = X.shape[0]
                te = round(n * test_size)
                torch.manual_seed(seed)
                perm = torch.randperm(n)
                tr_ids, te_ids = perm[: n - te], perm[n - te :]
                self.Xtr, self.ytr = X[tr_ids, :], y[tr_ids]
                self.Xte, self.yte = X[te_ids, :], y[te_ids]
            else:
                from sklearn.model_selection import train_test_split

                self.Xtr, self.Xte, self.ytr, self.yte = train_test_split(
                    X, y, test_size=test_size, random_state=seed, shuffle=True
                )
        else:
            self.Xtr, self.ytr = X, y
            self.Xte, self.yte = X_te, y_te

        self.n_classes = len(np.unique(self.ytr))

    def __getitem__(self, idx: Union[int, List[int]]) -> Tuple[Any, Any]:
        return self.Xtr[idx, :], self.ytr[idx]

    def at(self, idx: Union[int, List[int]], eval_set: bool = False) -> Optional[Tuple[Any, Any]]:
        """Sample(s) at ``idx``; returns ``None`` for an empty eval request."""
        if eval_set:
            if not isinstance(idx, (list, np.ndarray)) or len(idx):
                return self.Xte[idx, :], self.yte[idx]
            return None
        return self[idx]

    def size(self, dim: int = 0) -> int:
        return self.Xtr.shape[dim]

    def get_train_set(self) -> Tuple[Any, Any]:
        return self.Xtr, self.ytr

    def get_eval_set(self) -> Tuple[Any, Any]:
        return self.Xte, self.yte

    def eval_size(self) -> int:
        return self.Xte.shape[0] if self.Xte is not None else 0

    def __repr__(self) -> str:
        return (
            f"{self.__class__.__name__}(size_tr={self.size()}, "
            f"size_te={self.eval_size()}, n_feats={self.size(1)}, "
            f"n_classes={self.n_classes})"
        )


class ClusteringDataHandler(ClassificationDataHandler):
    """Unsupervised variant: the evaluation set *is* the training set
    (gossipy/data/handler.py:138-164)."""

    def __init__(self, X: Union[np.ndarray, torch.Tensor], y: Union[np.ndarray, torch.Tensor]):
        super().__init__(X, y, test_size=0)

    def get_eval_set(self) -> Tuple[Any, Any]:
        return self.get_train_set()

    def eval_size(self) -> int:
        return self.size()

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}(size={self.size()})"


class RegressionDataHandler(ClassificationDataHandler):
    """Float-label variant (gossipy/data/handler.py:168-178; the reference's
    ``at`` is missing its return — fixed here)."""

    def at(self, idx, eval_set: bool = False):
        return super().at(idx, eval_set)


class RecSysDataHandler(DataHandler):
    """Per-user rating lists with a per-user train/test split
    (gossipy/data/handler.py:181-245)."""

    def __init__(
        self,
        ratings: Dict[int, List[Tuple[int, float]]],
        n_users: int,
        n_items: int,
        test_size: float = 0.2,
        seed: int = 42,
    ):
        self.ratings = ratings
        self.n_users = n_users
        self.n_items = n_items
        self.test_id: List[int] = []
        np.random.seed(seed)
        for u in range(len(self.ratings)):
            self.test_id.append(max(1, int(len(self.ratings[u]) * (1 - test_size))))
            self.ratings[u] = np.random.permutation(self.ratings[u])

    def __getitem__(self, idx: int) -> List[Tuple[int, float]]:
        return self.ratings[idx][: self.test_id[idx]]

    def at(self, idx: int, eval_set: bool = False) -> List[Tuple[int, float]]:
        if eval_set:
            return self.ratings[idx][self.test_id[idx] :]
        return self[idx]

    def size(self, dim: int = 0) -> int:
        return self.n_users

    def get_train_set(self) -> Dict[int, Any]:
        return {u: self[u] for u in range(self.n_users)}

    def get_eval_set(self) -> Dict[int, Any]:
        return {u: self.at(u, True) for u in range(self.n_users)}

    def eval_size(self) -> int:
        return 0

    def __repr__(self) -> str:
        n_rat = sum(len(self.ratings[u]) for u in range(self.n_users))
        return (
            f"{self.__class__.__name__}(n_users={self.size()}, "
            f"n_items={self.n_items}, n_ratings={n_rat})"
        )
