"""Unit tests for the data layer: handlers, dispatcher, non-IID partitioners,
synthetic generators."""

import numpy as np
import pytest
import torch

from gossipy_amd import set_seed
from gossipy_amd.data import (
    AssignmentHandler,
    DataDispatcher,
    RecSysDataDispatcher,
    make_synthetic_classification,
    make_synthetic_recsys,
)
from gossipy_amd.data.handler import (
    ClassificationDataHandler,
    ClusteringDataHandler,
    RecSysDataHandler,
    RegressionDataHandler,
)


@pytest.fixture
def clf_handler():
    X, y = make_synthetic_classification((100, 5, 3), seed=0)
    return ClassificationDataHandler(X, y, test_size=0.2, seed=0)


class TestClassificationDataHandler:
    def test_split_sizes(self, clf_handler):
        assert clf_handler.size() == 80
        assert clf_handler.eval_size() == 20

    def test_at_train_and_eval(self, clf_handler):
        xb, yb = clf_handler.at([0, 1, 2])
        assert xb.shape == (3, 5) and yb.shape == (3,)
        xe, ye = clf_handler.at([0], eval_set=True)
        assert xe.shape == (1, 5)

    def test_explicit_eval_set(self):
        Xtr, ytr = make_synthetic_classification((40, 4, 2), seed=1)
        Xte, yte = make_synthetic_classification((10, 4, 2), seed=2)
        h = ClassificationDataHandler(Xtr, ytr, X_te=Xte, y_te=yte)
        assert h.size() == 40 and h.eval_size() == 10

    def test_seeded_split_reproducible(self):
        X, y = make_synthetic_classification((50, 4, 2), seed=3)
        h1 = ClassificationDataHandler(X, y, test_size=0.3, seed=5)
        h2 = ClassificationDataHandler(X, y, test_size=0.3, seed=5)
        a, b = h1.get_train_set(), h2.get_train_set()
        assert torch.equal(a[0], b[0]) and torch.equal(a[1], b[1])


class TestClusteringRegression:
    def test_clustering_eval_is_train(self):
        X, y = make_synthetic_classification((30, 3, 2), seed=0)
        h = ClusteringDataHandler(X, y)
        tr, ev = h.get_train_set(), h.get_eval_set()
        assert torch.equal(tr[0], ev[0])
        assert h.eval_size() == h.size()

    def test_regression_at_returns(self):
        X = torch.randn(20, 3)
        y = torch.randn(20)
        h = RegressionDataHandler(X, y, test_size=0.0)
        out = h.at([0, 1])
        # bug-fix parity check: the reference's RegressionDataHandler.at
        # forgets its return (gossipy/data/handler.py:175-178)
        assert out is not None


class TestRecSys:
    def test_handler_split(self):
        ratings, _, _ = make_synthetic_recsys(n_users=10, n_items=20, seed=0)
        h = RecSysDataHandler(ratings, 10, 20, test_size=0.2, seed=0)
        tr = h.at(0)
        te = h.at(0, eval_set=True)
        assert len(tr) > 0 and len(te) >= 0
        assert len(tr) + len(te) == len(ratings[0])

    def test_dispatcher_permutes_users(self):
        ratings, _, _ = make_synthetic_recsys(n_users=8, n_items=10, seed=0)
        h = RecSysDataHandler(ratings, 8, 10, test_size=0.2, seed=0)
        d = RecSysDataDispatcher(h)
        d.assign(seed=1)
        assert d.size() == 8
        assert not d.has_test()


class TestDataDispatcher:
    def test_uniform_assignment_covers(self, clf_handler):
        d = DataDispatcher(clf_handler, n=10, eval_on_user=True)
        tot = sum(len(d[i][0][0]) for i in range(10))
        assert tot == clf_handler.size()
        assert d.size() == 10

    def test_eval_on_user_splits_test(self, clf_handler):
        d = DataDispatcher(clf_handler, n=5, eval_on_user=True)
        tr, te = d[0]
        assert te is not None

    def test_no_eval_on_user(self, clf_handler):
        d = DataDispatcher(clf_handler, n=5, eval_on_user=False)
        tr, te = d[0]
        assert te is None
        assert d.has_test()

    def test_custom_assignments(self, clf_handler):
        d = DataDispatcher(clf_handler, n=2, eval_on_user=False, auto_assign=False)
        d.set_assignments([list(range(0, 40)), list(range(40, 80))])
        assert len(d[0][0][0]) == 40


class TestAssignmentHandler:
    def setup_method(self):
        set_seed(0)
        self.y = np.random.randint(0, 4, size=400)

    def _check_cover(self, assignments, n):
        assert len(assignments) == n
        all_ids = np.concatenate(assignments)
        assert len(np.unique(all_ids)) == len(all_ids), "no duplicates"
        return all_ids

    def test_uniform(self):
        a = AssignmentHandler(seed=0).uniform(self.y, 8)
        ids = self._check_cover(a, 8)
        assert len(ids) == 400

    def test_quantity_skew(self):
        a = AssignmentHandler(seed=0).quantity_skew(self.y, 8, min_quantity=2)
        self._check_cover(a, 8)
        assert all(len(x) >= 2 for x in a)

    def test_classwise_quantity_skew(self):
        a = AssignmentHandler(seed=0).classwise_quantity_skew(self.y, 8)
        self._check_cover(a, 8)

    def test_label_quantity_skew(self):
        a = AssignmentHandler(seed=0).label_quantity_skew(self.y, 8, class_per_client=2)
        self._check_cover(a, 8)
        for ids in a:
            assert len(np.unique(self.y[ids])) <= 2

    def test_label_dirichlet_skew(self):
        a = AssignmentHandler(seed=0).label_dirichlet_skew(self.y, 8, beta=0.5)
        self._check_cover(a, 8)

    def test_label_pathological_skew(self):
        a = AssignmentHandler(seed=0).label_pathological_skew(self.y, 8, shards_per_client=2)
        ids = self._check_cover(a, 8)
        assert len(ids) == 400


class TestSynthetic:
    def test_classification_shapes(self):
        X, y = make_synthetic_classification((64, 7, 3), seed=0)
        assert X.shape == (64, 7) and y.shape == (64,)
        assert y.max() < 3

    def test_named_shape(self):
        X, y = make_synthetic_classification("spambase", seed=0)
        assert X.shape[1] == 57

    def test_learnable(self):
        X, y = make_synthetic_classification((200, 10, 2), seed=0, margin=3.0)
        from sklearn.linear_model import LogisticRegression as SkLR

        clf = SkLR(max_iter=200).fit(X.numpy(), y.numpy())
        assert clf.score(X.numpy(), y.numpy()) > 0.9

    def test_recsys_ratings(self):
        r, _, _ = make_synthetic_recsys(n_users=5, n_items=12, seed=0)
        assert set(r.keys()) == set(range(5))
        for u, lst in r.items():
            assert all(0 <= i < 12 and 1 <= v <= 5 for i, v in lst)


def test_sklearn_builtin_loaders_offline():
    """The sklearn-bundled datasets need no network (parity with
    gossipy/data/__init__.py:561-624 for the offline subset)."""
    from gossipy_amd.data import load_classification_dataset

    for name, shape in (("iris", (150, 4)), ("wine", (178, 13))):
        X, y = load_classification_dataset(name, as_tensor=True)
        assert tuple(X.shape) == shape
        assert len(y) == shape[0]
