"""Unit tests for model handlers: training semantics, merge math, caching,
evaluation metrics, sampling and partition merges."""

import copy

import numpy as np
import pytest
import torch

from gossipy_amd import CACHE, set_seed
from gossipy_amd.core import CreateModelMode
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.model.handler import (
    AdaLineHandler,
    KMeansHandler,
    LimitedMergeTMH,
    MFModelHandler,
    PartitionedTMH,
    PegasosHandler,
    SamplingTMH,
    TorchModelHandler,
    WeightedTMH,
)
from gossipy_amd.model.nn import AdaLine, LogisticRegression, TorchMLP
from gossipy_amd.model.sampling import TorchModelPartition, TorchModelSampling
from gossipy_amd.utils import torch_models_eq


def _logreg_handler(din=8, dout=2, mode=CreateModelMode.MERGE_UPDATE, **kw):
    return TorchModelHandler(
        net=LogisticRegression(din, dout),
        optimizer=torch.optim.SGD,
        optimizer_params={"lr": 0.1},
        criterion=torch.nn.CrossEntropyLoss(),
        create_model_mode=mode,
        **kw,
    )


def _binary_data(n=64, d=8, seed=1):
    set_seed(seed)
    X, y = make_synthetic_classification((n, d, 2), seed=seed)
    return X, y


class TestTorchModelHandler:
    def test_update_increments_age_and_changes_params(self):
        h = _logreg_handler()
        h.init()
        X, y = _binary_data()
        before = copy.deepcopy(h.model.state_dict())
        h._update((X, y))
        assert h.n_updates > 0
        changed = any(
            not torch.equal(before[k], h.model.state_dict()[k]) for k in before
        )
        assert changed

    def test_merge_is_elementwise_mean(self):
        h1, h2 = _logreg_handler(), _logreg_handler()
        h1.init()
        h2.init()
        with torch.no_grad():
            for p in h1.model.parameters():
                p.fill_(1.0)
            for p in h2.model.parameters():
                p.fill_(3.0)
        h1._merge(h2)
        for p in h1.model.parameters():
            assert torch.allclose(p, torch.full_like(p, 2.0))

    def test_merge_many_is_mean_over_k_plus_1(self):
        hs = [_logreg_handler() for _ in range(3)]
        for i, h in enumerate(hs):
            h.init()
            with torch.no_grad():
                for p in h.model.parameters():
                    p.fill_(float(i))  # 0, 1, 2
        hs[0]._merge(hs[1:])
        for p in hs[0].model.parameters():
            assert torch.allclose(p, torch.full_like(p, 1.0))  # (0+1+2)/3

    def test_merge_takes_max_age(self):
        h1, h2 = _logreg_handler(), _logreg_handler()
        h1.init()
        h2.init()
        h1.n_updates, h2.n_updates = 3, 7
        h1._merge(h2)
        assert h1.n_updates == 7

    def test_call_update_mode_adopts_received(self):
        # UPDATE trains the RECEIVED model and adopts it
        # (reference quirk, gossipy/model/handler.py:122-125)
        h1 = _logreg_handler(mode=CreateModelMode.UPDATE)
        h2 = _logreg_handler(mode=CreateModelMode.UPDATE)
        h1.init()
        h2.init()
        X, y = _binary_data()
        h1(h2, (X, y))
        assert torch_models_eq(h1.model, h2.model)
        assert h1.n_updates == h2.n_updates

    def test_call_pass_mode_copies_model(self):
        h1 = _logreg_handler(mode=CreateModelMode.PASS)
        h2 = _logreg_handler()
        h1.init()
        h2.init()
        h1(h2, None)
        assert torch_models_eq(h1.model, h2.model)
        assert h1.model is not h2.model

    def test_evaluate_metrics_present_and_bounded(self):
        h = _logreg_handler()
        h.init()
        set_seed(1)
        X, y = make_synthetic_classification((64, 8, 2), seed=1, margin=3.0)
        for _ in range(20):
            h._update((X, y))
        res = h.evaluate((X, y))
        for k in ("accuracy", "precision", "recall", "f1_score", "auc"):
            assert k in res
            assert 0.0 <= res[k] <= 1.0
        assert res["accuracy"] > 0.8, "training on its own data should fit"

    def test_caching_pushes_snapshot(self):
        h = _logreg_handler()
        h.init()
        key = h.caching(owner=5)
        assert key.get() == (5, h.n_updates)
        snap = CACHE.pop(key)
        assert torch_models_eq(snap.model, h.model)
        # snapshot is independent of the live handler
        X, y = _binary_data()
        h._update((X, y))
        assert not torch_models_eq(snap.model, h.model)

    def test_copy_shares_stateless_criterion(self):
        h = _logreg_handler()
        h.init()
        c = h.copy()
        assert c.criterion is h.criterion  # the anti-deepcopy optimization
        assert torch_models_eq(c.model, h.model)
        assert c.model is not h.model

    def test_get_size_is_param_count(self):
        h = _logreg_handler(din=8, dout=2)
        h.init()
        assert h.get_size() == 8 * 2 + 2


class TestPegasosAdaLine:
    def test_adaline_delta_rule(self):
        h = AdaLineHandler(AdaLine(3), learning_rate=0.5)
        h.init()
        x = torch.tensor([[1.0, 2.0, 3.0]])
        y = torch.tensor([1.0])
        h._update((x, y))
        # w starts at 0 -> err = 1 -> w = 0.5 * 1 * x
        assert torch.allclose(h.model.model.data, 0.5 * x[0])
        assert h.n_updates == 1

    def test_pegasos_learns_separable(self):
        set_seed(3)
        X, y = make_synthetic_classification((200, 10, 2), seed=3, margin=3.0)
        y = 2 * y.float() - 1
        h = PegasosHandler(AdaLine(10), learning_rate=0.01)
        h.init()
        for _ in range(5):
            h._update((X, y))
        res = h.evaluate((X, y))
        assert res["accuracy"] > 0.9

    def test_merge_is_average(self):
        h1 = AdaLineHandler(AdaLine(2), 0.1)
        h2 = AdaLineHandler(AdaLine(2), 0.1)
        h1.init()
        h2.init()
        with torch.no_grad():
            h1.model.model += torch.tensor([2.0, 4.0])
            h2.model.model += torch.tensor([4.0, 8.0])
        h1._merge(h2)
        assert torch.allclose(h1.model.model.data, torch.tensor([3.0, 6.0]))


class TestSamplingAndPartition:
    def test_sample_respects_fraction(self):
        net = TorchMLP(10, 2, (16,))
        sample = TorchModelSampling.sample(0.3, net)
        total = sum(len(v[0]) for v in sample.values() if v is not None)
        assert total == max(1, round(0.3 * net.get_size()))

    def test_sample_merge_touches_only_sample(self):
        set_seed(0)
        net1, net2 = LogisticRegression(6, 2), LogisticRegression(6, 2)
        with torch.no_grad():
            for p in net1.parameters():
                p.fill_(0.0)
            for p in net2.parameters():
                p.fill_(2.0)
        sample = TorchModelSampling.sample(0.5, net1)
        TorchModelSampling.merge(sample, net1, net2)
        flat = torch.cat([p.detach().flatten() for p in net1.parameters()])
        # merged coords become 1.0, untouched stay 0.0
        assert set(np.round(flat.numpy(), 6)) <= {0.0, 1.0}
        assert (flat == 1.0).sum() > 0

    def test_partition_covers_all_params_exactly_once(self):
        net = TorchMLP(7, 3, (5,))
        part = TorchModelPartition(net, 4)
        seen = {}
        for p, per_layer in part.partitions.items():
            for li, ids in per_layer.items():
                if ids is None:
                    continue
                for coords in zip(*(t.tolist() for t in ids)):
                    key = (li, coords)
                    assert key not in seen, f"{key} in two partitions"
                    seen[key] = p
        assert len(seen) == net.get_size()

    def test_partition_sizes_balanced(self):
        net = LogisticRegression(7, 3)  # 24 params
        part = TorchModelPartition(net, 5)
        sizes = [hi - lo for lo, hi in part.flat_ranges()]
        assert sum(sizes) == net.get_size()
        assert max(sizes) - min(sizes) <= 1

    def test_partition_merge_weighted(self):
        net1, net2 = LogisticRegression(4, 2), LogisticRegression(4, 2)
        with torch.no_grad():
            for p in net1.parameters():
                p.fill_(0.0)
            for p in net2.parameters():
                p.fill_(4.0)
        part = TorchModelPartition(net1, 2)
        part.merge(0, net1, net2, weights=(1, 3))
        flat = torch.cat([p.detach().flatten() for p in net1.parameters()])
        merged = (flat == 3.0).sum().item()  # 0*1/4 + 4*3/4
        untouched = (flat == 0.0).sum().item()
        assert merged > 0 and untouched > 0
        assert merged + untouched == net1.get_size()


class TestPartitionedTMH:
    def _handler(self, n_parts=3):
        net = LogisticRegression(6, 2)
        return PartitionedTMH(
            net=net,
            tm_partition=TorchModelPartition(net, n_parts),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )

    def test_age_vector_increments_whole(self):
        h = self._handler()
        h.init()
        X, y = _binary_data(n=32, d=6)
        h._update((X, y))
        # quirk parity: every partition ages by 1 per batch
        assert (h.n_updates == h.n_updates[0]).all()
        assert h.n_updates[0] >= 1

    def test_caching_keys_on_age_vector_string(self):
        h = self._handler()
        h.init()
        key = h.caching(owner=2)
        assert key.get() == (2, str(h.n_updates))
        CACHE.pop(key)

    def test_partition_merge_updates_only_one_part_age(self):
        h1, h2 = self._handler(), self._handler()
        h1.init()
        h2.init()
        h1.n_updates = np.array([1, 1, 1])
        h2.n_updates = np.array([5, 5, 5])
        h1._merge(h2, id_part=1)
        assert list(h1.n_updates) == [1, 5, 1]


class TestWeightedAndLimited:
    def test_weighted_merge(self):
        hs = [_logreg_handler() for _ in range(3)]
        for i, h in enumerate(hs):
            h.init()
            with torch.no_grad():
                for p in h.model.parameters():
                    p.fill_(float(i + 1))  # 1, 2, 3
        wh = WeightedTMH(
            net=LogisticRegression(8, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )
        wh.init()
        with torch.no_grad():
            for p in wh.model.parameters():
                p.fill_(10.0)
        wh._merge(hs, weights=[0.4, 0.2, 0.2, 0.2])
        for p in wh.model.parameters():
            assert torch.allclose(p, torch.full_like(p, 10 * 0.4 + (1 + 2 + 3) * 0.2))

    def test_limited_merge_keeps_newer_local(self):
        h = LimitedMergeTMH(
            net=LogisticRegression(4, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            age_diff_threshold=2,
        )
        h.init()
        other = _logreg_handler(din=4)
        other.init()
        h.n_updates, other.n_updates = 10, 3
        before = copy.deepcopy(h.model.state_dict())
        h._merge(other)
        for k in before:
            assert torch.equal(before[k], h.model.state_dict()[k])

    def test_limited_merge_adopts_much_newer_remote(self):
        h = LimitedMergeTMH(
            net=LogisticRegression(4, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            age_diff_threshold=2,
        )
        h.init()
        other = _logreg_handler(din=4)
        other.init()
        h.n_updates, other.n_updates = 1, 10
        h._merge(other)
        assert torch_models_eq(h.model, other.model)
        assert h.n_updates == 10

    def test_limited_merge_age_weighted_within_threshold(self):
        h = LimitedMergeTMH(
            net=LogisticRegression(4, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
            age_diff_threshold=5,
        )
        h.init()
        other = _logreg_handler(din=4)
        other.init()
        with torch.no_grad():
            for p in h.model.parameters():
                p.fill_(0.0)
            for p in other.model.parameters():
                p.fill_(6.0)
        h.n_updates, other.n_updates = 1, 2
        h._merge(other)
        for p in h.model.parameters():
            assert torch.allclose(p, torch.full_like(p, 6.0 * 2 / 3))


class TestMFHandler:
    def test_update_reduces_rmse(self):
        set_seed(0)
        h = MFModelHandler(dim=4, n_items=20, learning_rate=0.05)
        h.init()
        ratings = [(i, float(1 + (i % 5))) for i in range(20)]
        before = h.evaluate(ratings)["rmse"]
        for _ in range(50):
            h._update(ratings)
        after = h.evaluate(ratings)["rmse"]
        assert after < before

    def test_merge_only_item_side(self):
        h1 = MFModelHandler(dim=2, n_items=5)
        h2 = MFModelHandler(dim=2, n_items=5)
        h1.init()
        h2.init()
        X1_before = h1.model[0][0].copy()
        h1._merge(h2)
        assert np.allclose(h1.model[0][0], X1_before), "user side must not merge"

    def test_get_size(self):
        h = MFModelHandler(dim=3, n_items=7)
        assert h.get_size() == 3 * 8


class TestKMeansHandler:
    def test_update_moves_centroids(self):
        set_seed(0)
        h = KMeansHandler(k=2, dim=2, alpha=0.5)
        h.init()
        X = torch.tensor([[10.0, 10.0], [-10.0, -10.0]])
        before = h.model.clone()
        h._update((X, None))
        assert not torch.equal(before, h.model)

    def test_naive_merge_is_mean(self):
        h1, h2 = KMeansHandler(2, 2), KMeansHandler(2, 2)
        h1.init()
        h2.init()
        h1.model = torch.zeros(2, 2)
        h2.model = torch.full((2, 2), 2.0)
        h1._merge(h2)
        assert torch.allclose(h1.model, torch.ones(2, 2))

    def test_hungarian_merge_matches_permuted(self):
        h1 = KMeansHandler(2, 2, matching="hungarian")
        h2 = KMeansHandler(2, 2, matching="hungarian")
        h1.init()
        h2.init()
        h1.model = torch.tensor([[0.0, 0.0], [10.0, 10.0]])
        h2.model = torch.tensor([[10.0, 10.0], [0.0, 0.0]])  # permuted
        h1._merge(h2)
        assert torch.allclose(h1.model, torch.tensor([[0.0, 0.0], [10.0, 10.0]]))

    def test_evaluate_nmi(self):
        set_seed(1)
        h = KMeansHandler(k=2, dim=2, alpha=0.2)
        h.init()
        X = torch.cat([torch.randn(50, 2) + 5, torch.randn(50, 2) - 5])
        y = torch.cat([torch.zeros(50), torch.ones(50)]).long()
        for _ in range(30):
            h._update((X, None))
        res = h.evaluate((X, y))
        assert res["nmi"] > 0.8


class TestSamplingTMH:
    def test_call_merge_update_with_sample(self):
        h1 = SamplingTMH(
            sample_size=0.5,
            net=LogisticRegression(6, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )
        h2 = SamplingTMH(
            sample_size=0.5,
            net=LogisticRegression(6, 2),
            optimizer=torch.optim.SGD,
            optimizer_params={"lr": 0.1},
            criterion=torch.nn.CrossEntropyLoss(),
        )
        h1.init()
        h2.init()
        X, y = _binary_data(n=32, d=6)
        sample = TorchModelSampling.sample(0.5, h2.model)
        h1(h2, (X, y), sample)
        assert h1.n_updates > 0


def test_perceptron_and_linear_regression_models():
    """Model-zoo parity: TorchPerceptron (gossipy/model/nn.py:26-64) and
    LinearRegression (:176-198) forward shapes + trainability."""
    import torch

    from gossipy_amd.model.nn import LinearRegression, TorchPerceptron

    p = TorchPerceptron(8)
    p.init_weights()
    out = p(torch.randn(5, 8))
    assert out.shape == (5, 1)
    assert (out >= 0).all() and (out <= 1).all()  # sigmoid head

    lr = LinearRegression(8, 1)
    x = torch.randn(64, 8)
    w = torch.randn(8, 1)
    y = x @ w
    opt = torch.optim.SGD(lr.parameters(), lr=0.05)
    for _ in range(300):
        opt.zero_grad()
        loss = torch.nn.functional.mse_loss(lr(x), y)
        loss.backward()
        opt.step()
    assert loss.item() < 0.05
    assert lr.get_size() == 9
