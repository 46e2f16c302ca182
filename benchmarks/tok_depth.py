"""Tokenized-round causal-depth analysis.

Measures, for the Hegedus-2021 tokenized+partitioned config, (a) how many
launch groups the entry-level packer emits per round and (b) the CAUSAL
DEPTH of the round's event graph — the longest snapshot->delivery
dependency chain, i.e. the number of sequential kernel launches ANY
scheduler must execute. Result (steady-state rounds):

    round 15: groups=32  deliveries=149  causal_depth=62
    round 20: groups=7   deliveries=32   causal_depth=13
    round 25: groups=14  deliveries=69   causal_depth=26
    round 30: groups=18  deliveries=93   causal_depth=35

groups ~ depth/2 everywhere: each group already covers two launch levels,
so the packer is AT the structural floor — a "true l4+ launch list"
cannot reduce the sequential launch count below ~depth/2 deliver launches.
The reactive burst chains (every delivery triggers a same-tick reactive
send under utility=1) make the round's work inherently sequential; the
remaining lever is per-dispatch latency of the tiny tick kernels
(~25 us measured), not packing.

Usage: python benchmarks/tok_depth.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from gossipy_amd.core import AntiEntropyProtocol, CreateModelMode
from gossipy_amd.data import make_synthetic_classification
from gossipy_amd.engine import (
    BatchedTokenizedGossipSimulator,
    DataArena,
    EngineConfig,
    LogRegSpec,
)
from gossipy_amd.flow_control import RandomizedTokenAccount
from examples.main_hegedus_2021 import k_regular_csr


def main():
    X, y = make_synthetic_classification((4600, 57, 2), seed=42)
    idx = np.random.default_rng(42).permutation(4600)
    cut = int(0.9 * 4600)
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], 100)]
    data = DataArena.from_shards(
        shards, torch.device("cpu"), global_eval=(X[idx[cut:]], y[idx[cut:]])
    )
    indptr, indices = k_regular_csr(100, 20)
    cfg = EngineConfig(
        n_nodes=100, delta=100, protocol=AntiEntropyProtocol.PUSH,
        model_size=116, sampling_eval=0.1, seed=42, n_parts=4,
        peers_indptr=indptr, peers_indices=indices,
    )
    spec = LogRegSpec(
        d_in=57, n_classes=2, lr=0.1, n_parts=4, mode=CreateModelMode.UPDATE
    )
    sim = BatchedTokenizedGossipSimulator(
        cfg, spec, data, token_account=RandomizedTokenAccount(C=20, A=10),
        utility_fun=lambda recv, sender, t: 1,
    )
    sim.init_nodes()
    for r in range(35):
        sched = sim.scheduler.next_round(r)
        if r < 12 or r % 5:
            continue
        f = sim._flatten_phases(sched.ticks)
        packed = sim._pack_flat(f)
        ng = len(packed["snap_tptr"]) - 1
        st, rt = f["snap_tptr"], f["recv_tptr"]
        nptr = f["recv_nptr"]
        delta = len(st) - 1
        slot_level, node_level, maxlev = {}, {}, 0
        for t in range(delta):
            for i in range(st[t], st[t + 1]):
                slot_level[f["snap_slots"][i]] = node_level.get(
                    f["snap_nodes"][i], 0
                )
            for i in range(rt[t], rt[t + 1]):
                n_ = f["recv_nodes"][i]
                for j in range(nptr[i], nptr[i + 1]):
                    lev = slot_level.get(f["del_slots"][j], 0) + 1
                    node_level[n_] = max(node_level.get(n_, 0), lev)
                    rs = f["reply_slots"][j]
                    if rs >= 0:
                        slot_level[rs] = node_level[n_]
                maxlev = max(maxlev, node_level[n_])
        print(
            f"round {r}: groups={ng} deliveries={len(f['del_slots'])} "
            f"causal_depth={maxlev}"
        )


if __name__ == "__main__":
    main()
