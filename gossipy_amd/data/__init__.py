"""Datasets, non-IID partitioners and dispatchers.

Parity layer for the reference's ``gossipy/data/__init__.py`` plus offline
synthetic generators (the MI355X target environment has no network egress, so
benchmarks run on synthetic data of the published datasets' shapes).
"""

from __future__ import annotations

import os
import shutil
from abc import ABC, abstractmethod
from typing import Any, Dict, List, Optional, Tuple, Union

import numpy as np
import torch
from numpy.random import choice, dirichlet, permutation, power, randint, shuffle
from torch import Tensor

from .. import LOG
from ..utils import download_and_untar, download_and_unzip

__all__ = [
    "DataHandler",
    "AssignmentHandler",
    "DataDispatcher",
    "RecSysDataDispatcher",
    "load_classification_dataset",
    "load_recsys_dataset",
    "get_CIFAR10",
    "get_FashionMNIST",
    "get_FEMNIST",
    "make_synthetic_classification",
    "make_synthetic_recsys",
]

#: shapes of the UCI datasets the reference downloads
#: (gossipy/data/__init__.py:43-52) — used to size synthetic stand-ins.
SYNTHETIC_SHAPES = {
    "spambase": (4601, 57, 2),
    "sonar": (208, 60, 2),
    "ionosphere": (351, 34, 2),
    "banknote": (1372, 4, 2),
}


class DataHandler(ABC):
    """Dataset interface (gossipy/data/__init__.py:55-161)."""

    @abstractmethod
    def __getitem__(self, idx: Union[int, List[int]]) -> Any:
        """Training sample(s) at ``idx``."""
        raise NotImplementedError

    @abstractmethod
    def at(self, idx: Union[int, List[int]], eval_set: bool = False) -> Any:
        """Sample(s) at ``idx`` from the training or evaluation set."""
        raise NotImplementedError

    @abstractmethod
    def size(self, dim: int = 0) -> int:
        """Training-set size along ``dim``."""
        raise NotImplementedError

    @abstractmethod
    def get_eval_set(self) -> Tuple[Any, Any]:
        """The evaluation set."""
        raise NotImplementedError

    @abstractmethod
    def get_train_set(self) -> Tuple[Any, Any]:
        """The training set."""
        raise NotImplementedError

    @abstractmethod
    def eval_size(self) -> int:
        """Number of evaluation examples."""
        raise NotImplementedError


class AssignmentHandler:
    """Non-IID example->client assignment strategies
    (gossipy/data/__init__.py:164-373).

    All six reference strategies are provided: ``uniform``, ``quantity_skew``,
    ``classwise_quantity_skew``, ``label_quantity_skew``,
    ``label_dirichlet_skew``, ``label_pathological_skew``.
    """

    def __init__(self, seed: int):
        torch.manual_seed(seed)
        np.random.seed(seed)

    def uniform(self, y: Union[np.ndarray, torch.Tensor], n: int) -> List[np.ndarray]:
        """Shuffle and split evenly across ``n`` clients."""
        per_client = y.shape[0] // n
        idx = permutation(y.shape[0])
        return [idx[per_client * i : per_client * (i + 1)] for i in range(n)]

    def quantity_skew(
        self,
        y: Union[np.ndarray, torch.Tensor],
        n: int,
        min_quantity: int = 2,
        alpha: float = 4.0,
    ) -> List[np.ndarray]:
        """Power-law sized shards: ``P(x; a) = a x^{a-1}`` over client ids,
        with at least ``min_quantity`` examples per client."""
        assert min_quantity * n <= y.shape[0], "# of instances must be > than min_quantity*n"
        assert min_quantity > 0, "min_quantity must be >= 1"
        skewed = np.array(power(alpha, y.shape[0] - min_quantity * n) * n, dtype=int)
        guaranteed = np.repeat(np.arange(n), min_quantity)
        assignment = np.concatenate([skewed, guaranteed])
        shuffle(assignment)
        return [np.where(assignment == i)[0] for i in range(n)]

    def classwise_quantity_skew(
        self,
        y: Union[np.ndarray, torch.Tensor],
        n: int,
        min_quantity: int = 2,
        alpha: float = 4.0,
    ) -> List[np.ndarray]:
        """Per-class power-law assignment; every client gets at least one
        example of each class."""
        assert min_quantity * n <= y.shape[0], "# of instances must be > than min_quantity*n"
        assert min_quantity > 0, "min_quantity must be >= 1"
        y_np = y.numpy() if isinstance(y, torch.Tensor) else y
        labels = list(range(len(np.unique(y_np))))
        class_sizes = [np.where(y_np == c)[0].shape[0] for c in labels]
        assert min(class_sizes) >= n, "Under represented class!"

        assignment = []
        for c in labels:
            skewed = np.array(power(alpha, class_sizes[c] - n) * n, dtype=int)
            full = np.concatenate([skewed, np.arange(n)])
            shuffle(full)
            assignment.append(full)

        result: List[List[int]] = [[] for _ in range(n)]
        for c in labels:
            ids_c = np.where(y_np == c)[0]
            for i in range(n):
                result[i] += list(ids_c[np.where(assignment[c] == i)[0]])
        return [np.array(r, dtype=int) for r in result]

    def label_quantity_skew(
        self,
        y: Union[np.ndarray, torch.Tensor],
        n: int,
        class_per_client: int = 2,
    ) -> List[np.ndarray]:
        """Each client holds examples of exactly ``class_per_client`` labels
        (https://arxiv.org/pdf/2102.02079.pdf)."""
        y_np = y.numpy() if isinstance(y, torch.Tensor) else y
        labels = set(np.unique(y_np))
        assert 0 < class_per_client <= len(labels), "class_per_client must be > 0 and <= #classes"
        assert class_per_client * n >= len(labels), "class_per_client * n must be >= #classes"
        client_labels = [choice(len(labels), class_per_client, replace=False) for _ in range(n)]
        covered = set().union(*[set(a) for a in client_labels])
        while len(covered) < len(labels):
            for missing in labels - covered:
                client_labels[randint(0, n)][randint(0, class_per_client)] = missing
            covered = set().union(*[set(a) for a in client_labels])
        class_map = {
            c: [u for u, lbls in enumerate(client_labels) if c in lbls] for c in labels
        }
        assignment = np.zeros(y_np.shape[0])
        for lbl, users in class_map.items():
            ids = np.where(y_np == lbl)[0]
            assignment[ids] = choice(users, len(ids))
        return [np.where(assignment == i)[0] for i in range(n)]

    def label_dirichlet_skew(
        self, y: torch.Tensor, n: int, beta: float = 0.1
    ) -> List[np.ndarray]:
        """Sample ``p_k ~ Dir_n(beta)`` per class and allocate proportionally
        (https://arxiv.org/pdf/2102.02079.pdf); each client is guaranteed one
        example per class."""
        assert beta > 0, "beta must be > 0"
        y_np = y.numpy() if isinstance(y, torch.Tensor) else y
        labels = set(np.unique(y_np))
        pk = {c: dirichlet([beta] * n, size=1)[0] for c in labels}
        assignment = np.zeros(y_np.shape[0])
        for c in labels:
            ids = np.where(y_np == c)[0]
            shuffle(ids)
            shuffle(pk[c])
            assignment[ids[n:]] = choice(n, size=len(ids) - n, p=pk[c])
            assignment[ids[:n]] = list(range(n))
        return [np.where(assignment == i)[0] for i in range(n)]

    def label_pathological_skew(
        self,
        y: Union[np.ndarray, torch.Tensor],
        n: int,
        shards_per_client: int = 2,
    ) -> List[np.ndarray]:
        """Sort by label, shard, hand each client ``shards_per_client``
        shards (McMahan 2017)."""
        y_np = y.numpy() if isinstance(y, torch.Tensor) else y
        sorted_ids = np.argsort(y_np)
        n_shards = int(shards_per_client * n)
        shard_size = int(np.ceil(len(y_np) / n_shards))
        assignment = np.zeros(y_np.shape[0])
        perm = permutation(n_shards)
        j = 0
        for i in range(n):
            for _ in range(shards_per_client):
                left = perm[j] * shard_size
                right = min((perm[j] + 1) * shard_size, len(y_np))
                assignment[sorted_ids[left:right]] = i
                j += 1
        return [np.where(assignment == i)[0] for i in range(n)]


class DataDispatcher:
    """Assigns example-index lists to ``n`` clients
    (gossipy/data/__init__.py:376-510)."""

    def __init__(
        self,
        data_handler: DataHandler,
        n: int = 0,
        eval_on_user: bool = True,
        auto_assign: bool = True,
    ):
        assert data_handler.size() >= n
        if n <= 1:
            n = data_handler.size()
        self.data_handler = data_handler
        self.n = n
        self.eval_on_user = eval_on_user
        self.tr_assignments: Optional[List] = None
        self.te_assignments: Optional[List] = None
        if auto_assign:
            self.assign()

    def set_assignments(
        self, tr_assignments: List, te_assignments: Optional[List] = None
    ) -> None:
        """Install custom (e.g. non-IID) assignments."""
        assert len(tr_assignments) == self.n
        assert not te_assignments or len(te_assignments) == self.n
        self.tr_assignments = tr_assignments
        self.te_assignments = (
            te_assignments if te_assignments else [[] for _ in range(self.n)]
        )

    def assign(self, seed: Optional[int] = 42) -> None:
        """Uniform random assignment."""
        handler = AssignmentHandler(seed)
        self.tr_assignments = handler.uniform(self.data_handler.ytr, self.n)
        if self.eval_on_user:
            self.te_assignments = handler.uniform(self.data_handler.yte, self.n)
        else:
            self.te_assignments = [[] for _ in range(self.n)]

    def __getitem__(self, idx: int) -> Any:
        assert 0 <= idx < self.n, "Index %d out of range." % idx
        return (
            self.data_handler.at(self.tr_assignments[idx]),
            self.data_handler.at(self.te_assignments[idx], True),
        )

    def size(self) -> int:
        """Number of clients."""
        return self.n

    def get_eval_set(self) -> Tuple[Any, Any]:
        """The global test set."""
        return self.data_handler.get_eval_set()

    def has_test(self) -> bool:
        """Whether a global test set exists."""
        return self.data_handler.eval_size() > 0

    def __repr__(self) -> str:
        return "DataDispatcher(handler=%s, n=%d, eval_on_user=%s)" % (
            self.data_handler,
            self.n,
            self.eval_on_user,
        )


class RecSysDataDispatcher(DataDispatcher):
    """One user = one client (gossipy/data/__init__.py:513-558)."""

    def __init__(self, data_handler):
        # deliberately skips DataDispatcher.__init__, like the reference
        self.data_handler = data_handler
        self.n = data_handler.n_users
        self.eval_on_user = True
        self.assignments: Optional[List[int]] = None

    def assign(self, seed: int = 42) -> None:
        torch.manual_seed(seed)
        self.assignments = torch.randperm(self.data_handler.size()).tolist()

    def __getitem__(self, idx: int) -> Any:
        assert 0 <= idx < self.n, "Index %d out of range." % idx
        return (
            self.data_handler.at(self.assignments[idx]),
            self.data_handler.at(self.assignments[idx], True),
        )

    def size(self) -> int:
        return self.n

    def get_eval_set(self) -> Optional[Tuple[Any, Any]]:
        return None

    def has_test(self) -> bool:
        return False

    def __repr__(self) -> str:
        return f"RecSysDataDispatcher(handler={self.data_handler}, eval_on_user={self.eval_on_user})"


def make_synthetic_classification(
    name_or_shape: Union[str, Tuple[int, int, int]],
    seed: int = 42,
    normalize: bool = True,
    as_tensor: bool = True,
    margin: float = 1.0,
) -> Union[Tuple[torch.Tensor, torch.Tensor], Tuple[np.ndarray, np.ndarray]]:
    """Generate a learnable synthetic classification dataset.

    Sized like a named reference dataset (see :data:`SYNTHETIC_SHAPES`) or an
    explicit ``(n_samples, n_features, n_classes)``. Samples are Gaussian
    blobs around random class centroids (separation ``margin``), so gossip
    learning curves actually rise — used by the benchmarks, which have no
    network access to the real UCI data.
    """
    if isinstance(name_or_shape, str):
        shape = SYNTHETIC_SHAPES[name_or_shape]
    else:
        shape = name_or_shape
    n, d, k = shape
    rng = np.random.default_rng(seed)
    centroids = rng.normal(0.0, margin, size=(k, d))
    y = rng.integers(0, k, size=n)
    X = centroids[y] + rng.normal(0.0, 1.0, size=(n, d))
    if normalize:
        X = (X - X.mean(axis=0)) / (X.std(axis=0) + 1e-12)
    if as_tensor:
        return torch.tensor(X).float(), torch.tensor(y).long()
    return X, y


def make_synthetic_recsys(
    n_users: int,
    n_items: int,
    ratings_per_user: int = 100,
    k_latent: int = 5,
    seed: int = 42,
) -> Tuple[Dict[int, List[Tuple[int, float]]], int, int]:
    """Generate synthetic MovieLens-shaped ratings from a random low-rank
    model (ratings clipped to [1, 5])."""
    rng = np.random.default_rng(seed)
    U = rng.normal(0, 1, size=(n_users, k_latent))
    V = rng.normal(0, 1, size=(n_items, k_latent))
    ratings: Dict[int, List[Tuple[int, float]]] = {}
    for u in range(n_users):
        items = rng.choice(n_items, size=min(ratings_per_user, n_items), replace=False)
        raw = U[u] @ V[items].T
        scaled = np.clip(np.round(3.0 + 1.5 * raw / (np.std(raw) + 1e-9)), 1, 5)
        ratings[u] = [(int(i), float(r)) for i, r in zip(items, scaled)]
    return ratings, n_users, n_items


def load_classification_dataset(
    name_or_path: str,
    normalize: bool = True,
    as_tensor: bool = True,
) -> Union[Tuple[torch.Tensor, torch.Tensor], Tuple[np.ndarray, np.ndarray]]:
    """Load a classification dataset by name or svmlight path
    (gossipy/data/__init__.py:561-624).

    The sklearn built-ins (iris, breast, digits, wine) work offline. The UCI
    names (spambase, sonar, ionosphere, abalone, banknote) require network
    access in the reference; here they fall back to synthetic stand-ins of
    the same shape with a warning when the download fails.
    """
    from sklearn import datasets
    from sklearn.preprocessing import LabelEncoder, StandardScaler

    if name_or_path == "iris":
        ds = datasets.load_iris()
        X, y = ds.data, ds.target
    elif name_or_path == "breast":
        ds = datasets.load_breast_cancer()
        X, y = ds.data, ds.target
    elif name_or_path == "digits":
        ds = datasets.load_digits()
        X, y = ds.data, ds.target
    elif name_or_path == "wine":
        ds = datasets.load_wine()
        X, y = ds.data, ds.target
    elif name_or_path == "reuters":
        from sklearn.datasets import load_svmlight_file

        url = "http://download.joachims.org/svm_light/examples/example1.tar.gz"
        folder = download_and_untar(url)[0]
        X_tr, y_tr = load_svmlight_file(folder + "/train.dat")
        X_te, y_te = load_svmlight_file(folder + "/test.dat")
        X_te = np.pad(X_te.toarray(), [(0, 0), (0, 17)], mode="constant")
        X = np.vstack([X_tr.toarray(), X_te])
        y = LabelEncoder().fit_transform(np.concatenate([y_tr, y_te]))
        shutil.rmtree(folder)
    elif name_or_path in SYNTHETIC_SHAPES or name_or_path == "abalone":
        try:
            import pandas as pd

            base = "https://archive.ics.uci.edu/ml/machine-learning-databases/"
            urls = {
                "spambase": (base + "spambase/spambase.data", 57),
                "sonar": (base + "undocumented/connectionist-bench/sonar/sonar.all-data", 60),
                "ionosphere": (base + "ionosphere/ionosphere.data", 34),
                "abalone": (base + "abalone/abalone.data", 0),
                "banknote": (base + "00267/data_banknote_authentication.txt", 4),
            }
            url, label_col = urls[name_or_path]
            LOG.info("Downloading dataset %s from '%s'." % (name_or_path, url))
            data = pd.read_csv(url, header=None).to_numpy()
            y = LabelEncoder().fit_transform(data[:, label_col])
            X = np.delete(data, [label_col], axis=1).astype("float64")
        except Exception as e:  # offline environment
            if name_or_path not in SYNTHETIC_SHAPES:
                raise
            LOG.warning(
                "Download of %s failed (%s); generating a synthetic stand-in "
                "of the same shape." % (name_or_path, e)
            )
            return make_synthetic_classification(
                name_or_path, normalize=normalize, as_tensor=as_tensor
            )
    else:
        from sklearn.datasets import load_svmlight_file

        X, y = load_svmlight_file(name_or_path)
        X = X.toarray()

    if normalize:
        X = StandardScaler().fit_transform(X)
    if as_tensor:
        X = torch.tensor(X).float()
        y = torch.tensor(y).long()
    return X, y


def load_recsys_dataset(
    name: str, path: str = "."
) -> Tuple[Dict[int, List[Tuple[int, float]]], int, int]:
    """Load a MovieLens dataset (gossipy/data/__init__.py:628-681).

    Requires network access; use :func:`make_synthetic_recsys` offline.
    """
    if name not in {"ml-100k", "ml-1m", "ml-10m", "ml-20m"}:
        raise ValueError("Unknown dataset %s." % name)
    folder = download_and_unzip(
        "https://files.grouplens.org/datasets/movielens/%s.zip" % name
    )[0]
    filename, sep = {
        "ml-100k": ("u.data", "\t"),
        "ml-20m": ("ratings.csv", ","),
    }.get(name, ("ratings.dat", "::"))

    ratings: Dict[int, List[Tuple[int, float]]] = {}
    umap: Dict[int, int] = {}
    imap: Dict[int, int] = {}
    with open(os.path.join(path, folder, filename), "r") as f:
        for line in f:
            u_raw, i_raw, r_raw = line.strip().split(sep)[0:3]
            u, i, r = int(u_raw), int(i_raw), float(r_raw)
            if u not in umap:
                umap[u] = len(umap)
                ratings[umap[u]] = []
            if i not in imap:
                imap[i] = len(imap)
            ratings[umap[u]].append((imap[i], r))
    shutil.rmtree(folder)
    return ratings, len(umap), len(imap)


def _require_torchvision():
    try:
        import torchvision  # noqa: F401

        return torchvision
    except ImportError as e:
        raise ImportError(
            "torchvision is not installed in this environment; use "
            "make_synthetic_classification((n, 3*32*32, 10)) or the engine's "
            "synthetic CIFAR-shaped generators instead."
        ) from e


def get_CIFAR10(path: str = "./data", as_tensor: bool = True):
    """CIFAR10 via torchvision (gossipy/data/__init__.py:684-722); needs
    torchvision + network."""
    torchvision = _require_torchvision()
    from pathlib import Path

    download = not Path(os.path.join(path, "/cifar-10-batches-py")).is_dir()
    train = torchvision.datasets.CIFAR10(root=path, train=True, download=download)
    test = torchvision.datasets.CIFAR10(root=path, train=False, download=download)
    if as_tensor:
        return (
            (torch.tensor(train.data).float().permute(0, 3, 1, 2) / 255.0,
             torch.tensor(train.targets)),
            (torch.tensor(test.data).float().permute(0, 3, 1, 2) / 255.0,
             torch.tensor(test.targets)),
        )
    return (train.data, train.targets), (test.data, test.targets)


def get_FashionMNIST(path: str = "./data", as_tensor: bool = True):
    """FashionMNIST via torchvision (gossipy/data/__init__.py:725-762)."""
    torchvision = _require_torchvision()
    from pathlib import Path

    download = not Path(os.path.join(path, "/FashionMNIST/raw/")).is_dir()
    train = torchvision.datasets.FashionMNIST(root=path, train=True, download=download)
    test = torchvision.datasets.FashionMNIST(root=path, train=False, download=download)
    if as_tensor:
        return (train.data / 255.0, train.targets), (test.data / 255.0, test.targets)
    return (
        (train.data.numpy() / 255.0, train.targets.numpy()),
        (test.data.numpy() / 255.0, test.targets.numpy()),
    )


def get_FEMNIST(path: str = "./data"):
    """FEMNIST tarball with per-writer assignments
    (gossipy/data/__init__.py:765-778); needs network."""
    url = "https://raw.githubusercontent.com/tao-shen/FEMNIST_pytorch/master/femnist.tar.gz"
    te_name, tr_name = download_and_untar(url, path)
    Xtr, ytr, ids_tr = torch.load(os.path.join(path, tr_name))
    Xte, yte, ids_te = torch.load(os.path.join(path, te_name))
    tr_assignment, te_assignment = [], []
    sum_tr = sum_te = 0
    for i in range(len(ids_tr)):
        ntr, nte = ids_tr[i], ids_te[i]
        tr_assignment.append(list(range(sum_tr, sum_tr + ntr)))
        te_assignment.append(list(range(sum_te, sum_te + nte)))
        sum_tr += ntr
        sum_te += nte
    return (Xtr, ytr, tr_assignment), (Xte, yte, te_assignment)
