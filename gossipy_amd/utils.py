"""Utility helpers: RNG helpers, model equality, dataset fetch, plotting.

Parity layer for the reference's ``gossipy/utils.py``. Network downloads are
kept for API parity but the target environment has no egress — prefer the
synthetic generators in :mod:`gossipy_amd.data`.
"""

from __future__ import annotations

import tarfile
from io import BytesIO
from json import JSONEncoder
from typing import Dict, List
from urllib.error import URLError
from urllib.request import urlopen
from zipfile import ZipFile

import numpy as np
import torch
from numpy.random import randint

from . import LOG

__all__ = [
    "choice_not_n",
    "torch_models_eq",
    "download_and_unzip",
    "download_and_untar",
    "plot_evaluation",
    "StringEncoder",
]


def choice_not_n(mn: int, mx: int, notn: int) -> int:
    """Uniform integer in ``[mn, mx)`` excluding ``notn`` (gossipy/utils.py:41-64)."""
    c = randint(mn, mx)
    while c == notn:
        c = randint(mn, mx)
    return int(c)


def torch_models_eq(m1: torch.nn.Module, m2: torch.nn.Module) -> bool:
    """State-dict equality of two torch modules (gossipy/utils.py:67-95)."""
    sd1, sd2 = m1.state_dict(), m2.state_dict()
    if len(sd1) != len(sd2):
        return False
    for (k1, t1), (k2, t2) in zip(sd1.items(), sd2.items()):
        if k1 != k2 or not torch.equal(t1, t2):
            return False
    return True


def download_and_unzip(url: str, extract_to: str = ".") -> List[str]:
    """Fetch a zip from ``url`` and extract it (gossipy/utils.py:98-126).

    Unavailable in the offline target environment; raises URLError there.
    """
    LOG.info("Downloading %s into %s" % (url, extract_to))
    try:
        response = urlopen(url)
    except URLError:
        import ssl

        ssl._create_default_https_context = ssl._create_unverified_context
        response = urlopen(url)
    archive = ZipFile(BytesIO(response.read()))
    archive.extractall(path=extract_to)
    return archive.namelist()


def download_and_untar(url: str, extract_to: str = ".") -> List[str]:
    """Fetch a tarball from ``url`` and extract it (gossipy/utils.py:129-149)."""
    LOG.info("Downloading %s into %s" % (url, extract_to))
    stream = urlopen(url)
    archive = tarfile.open(fileobj=stream, mode="r|gz")
    archive.extractall(path=extract_to)
    return archive.getnames()


def plot_evaluation(evals: List[List[Dict]], title: str = "Untitled plot") -> None:
    """Plot mean±std metric curves over repeated runs (gossipy/utils.py:152-183).

    Imports matplotlib lazily so headless environments without it can still
    use the rest of the package.
    """
    if not evals or not evals[0] or not evals[0][0]:
        return
    # accept either the reference's list-of-dicts runs or
    # SimulationReport.get_evaluation's (round, dict) tuples
    evals = [
        [p[1] if isinstance(p, tuple) else p for p in run] for run in evals
    ]
    import matplotlib.pyplot as plt

    fig = plt.figure()
    try:
        fig.canvas.manager.set_window_title(title)
    except Exception:
        pass
    ax = fig.add_subplot(111)
    for metric in evals[0][0]:
        curves = [[point[metric] for point in run] for run in evals]
        mu = np.mean(curves, axis=0)
        std = np.std(curves, axis=0)
        xs = range(1, len(mu) + 1)
        plt.fill_between(xs, mu - std, mu + std, alpha=0.2)
        plt.plot(xs, mu, label=metric)
        LOG.info(f"{metric}: {mu[-1]:.2f}")
    plt.title(title)
    plt.xlabel("cycle")
    plt.ylabel("metric value")
    ax.legend(loc="lower right")
    plt.show()


class StringEncoder(JSONEncoder):
    """JSON encoder that stringifies anything non-serializable
    (gossipy/utils.py:186-189)."""

    def default(self, o) -> str:
        return str(o)
