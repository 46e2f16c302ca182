"""Flagship benchmark: Hegedus-2021 gossip learning, 1000 nodes per GPU.

Measures the BASELINE.json metric — gossip rounds/sec (whole node) on the
Hegedus-2021 logreg config (1000 nodes, LogReg 57x2, TorchModelHandler
MERGE_UPDATE, PUSH, full mesh, delta=100, sampling_eval=0.01, synthetic
spambase-shaped data) — on the batched MI355X engine. Weak scaling: each
GPU hosts 1000 simulated nodes; the whole-job aggregate reported as
``value`` is node-rounds/sec = total_nodes x rounds/sec.

Run single GPU:  python bench.py --gpus 1 --steps 50 --warmup 10
Multi GPU (driver): python -m torch.distributed.run --nnodes=1
    --nproc-per-node N --master-addr 127.0.0.1 bench.py --gpus N ...

The reference (measured on CPU) does 0.79 rounds/sec at 1000
nodes = 790 node-rounds/sec; ``vs_baseline`` is value / 790.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch

NODES_PER_GPU = 1000
D_IN, N_CLASSES = 57, 2
DELTA = 100
SAMPLES_PER_SHARD_SET = 4601  # spambase-shaped synthetic, per 1000 nodes
BASELINE_NODE_ROUNDS_PER_SEC = 0.79 * 1000


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=30, help="rounds to time")
    ap.add_argument("--warmup", type=int, default=5, help="untimed rounds")
    ap.add_argument("--nodes-per-gpu", type=int, default=NODES_PER_GPU)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    import torch.distributed as dist

    if world > 1:
        # device_id => eager (collective) communicator init; without it the
        # first NCCL op could be a pairwise batch_isend_irecv, whose lazy
        # world-comm init deadlocks ranks that have no op in that group
        if use_cuda:
            dist.init_process_group("nccl", device_id=device)
        else:
            dist.init_process_group("gloo")

    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.data import make_synthetic_classification
    from gossipy_amd.engine import (
        BatchedGossipSimulator,
        DataArena,
        EngineConfig,
        LogRegSpec,
    )
    from gossipy_amd.simul import SimulationReport

    n_nodes = args.nodes_per_gpu * world
    # dataset size tracks the TOTAL population (4601 samples per 1000
    # nodes), so a given (n_nodes, seed) pair sees identical data at any
    # GPU count — results are residency-invariant, only faster
    n_samples = max(1, SAMPLES_PER_SHARD_SET * n_nodes // NODES_PER_GPU)

    # deterministic global dataset; every rank generates it and slices its
    # residency block's shards (no network, random-init weights)
    X, y = make_synthetic_classification((n_samples, D_IN, N_CLASSES), seed=7)
    rng = np.random.default_rng(7)
    idx = rng.permutation(n_samples)
    n_eval = max(256, n_samples // 100)
    eval_idx, train_idx = idx[:n_eval], idx[n_eval:]
    shard_ids = np.array_split(train_idx, n_nodes)
    lo = rank * args.nodes_per_gpu
    my_shards = [(X[s], y[s]) for s in shard_ids[lo : lo + args.nodes_per_gpu]]
    data = DataArena.from_shards(
        my_shards, device, global_eval=(X[eval_idx], y[eval_idx])
    )

    cfg = EngineConfig(
        n_nodes=n_nodes,
        delta=DELTA,
        protocol=AntiEntropyProtocol.PUSH,
        model_size=N_CLASSES * D_IN + N_CLASSES,
        sampling_eval=0.01,
        seed=42,
        sync=True,
    )
    spec = LogRegSpec(
        d_in=D_IN, n_classes=N_CLASSES, lr=0.1, local_epochs=1, batch_size=32
    )
    sim = BatchedGossipSimulator(cfg, spec, data, device=device)
    report = SimulationReport()
    sim.add_receiver(report)
    sim.init_nodes()

    def barrier_sync():
        if world > 1:
            dist.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    sim.start(n_rounds=args.warmup)
    barrier_sync()
    t0 = time.perf_counter()
    sim.start(n_rounds=args.steps)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
    if world > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    rounds_per_sec = args.steps / elapsed
    value = rounds_per_sec * n_nodes  # whole-job aggregate: node-rounds/sec
    if rank == 0:
        acc = None
        evals = report.get_evaluation(False)
        if evals:
            acc = round(evals[-1][1].get("accuracy", float("nan")), 4)
        out = {
            "metric": "node-rounds/sec (gossip rounds/sec x nodes, Hegedus-2021 logreg)",
            "value": round(value, 2),
            "unit": "node-rounds/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000.0 * elapsed / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(value / BASELINE_NODE_ROUNDS_PER_SEC, 2),
            "dtype": "fp32",
            "data": f"synthetic spambase-shaped ({SAMPLES_PER_SHARD_SET}x{D_IN} per {NODES_PER_GPU} nodes), random-init weights",
            "config": {
                "model": "logreg-57x2 (Hegedus-2021)",
                "global_batch": None,
                "seq_len": None,
                "nodes": n_nodes,
                "nodes_per_gpu": args.nodes_per_gpu,
                "delta": DELTA,
                "protocol": "push",
                "mode": "merge_update",
                "sampling_eval": 0.01,
                "rounds_per_sec": round(rounds_per_sec, 3),
                "final_accuracy": acc,
                "parallelism": f"node-sharded gossip over RCCL/xGMI (dp{world})",
            },
        }
        print(json.dumps(out))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
