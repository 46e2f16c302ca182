"""Centralized baselines (reference baseline.py:11-97): a torch MLP and an
sklearn MLP trained on the pooled dataset, as oracles for the gossip
curves. Synthetic spambase-shaped data.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

import numpy as np
import torch
from sklearn.neural_network import MLPClassifier

from gossipy_amd.data import make_synthetic_classification


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--samples", type=int, default=4600)
    ap.add_argument("--epochs", type=int, default=30)
    args = ap.parse_args()

    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    X, y = make_synthetic_classification((args.samples, 57, 2), seed=42, margin=2.0)
    idx = np.random.default_rng(42).permutation(args.samples)
    cut = int(0.9 * args.samples)
    Xtr, ytr = X[idx[:cut]].to(device), y[idx[:cut]].long().to(device)
    Xte, yte = X[idx[cut:]].to(device), y[idx[cut:]].long().to(device)

    net = torch.nn.Sequential(
        torch.nn.Linear(57, 100), torch.nn.ReLU(), torch.nn.Linear(100, 2)
    ).to(device)
    opt = torch.optim.SGD(net.parameters(), lr=0.1)
    crit = torch.nn.CrossEntropyLoss()
    for _ in range(args.epochs):
        for s in range(0, len(Xtr), 32):
            opt.zero_grad()
            crit(net(Xtr[s : s + 32]), ytr[s : s + 32]).backward()
            opt.step()
    acc = float((net(Xte).argmax(1) == yte).float().mean())
    print(f"torch MLP accuracy:   {acc:.4f}")

    sk = MLPClassifier(hidden_layer_sizes=(100,), max_iter=args.epochs * 10)
    sk.fit(Xtr.cpu().numpy(), ytr.cpu().numpy())
    print(f"sklearn MLP accuracy: {sk.score(Xte.cpu().numpy(), yte.cpu().numpy()):.4f}")


if __name__ == "__main__":
    main()
