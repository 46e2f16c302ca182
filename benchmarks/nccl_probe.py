"""Probe: can RCCL run 2 ranks on ONE MI355X (both ranks cuda:0)?

If yes, the engine's real NCCL/RCCL exchange path (init_process_group
('nccl', device_id=...) + grouped batch_isend_irecv + the 2-rank engine
equivalence) can be exercised on a single-GPU box; if RCCL refuses
duplicate devices (stock NCCL behavior), we record that and rely on the
gloo 2/4/8-rank rehearsals + the driver's 8-GPU SCALE run.

Usage: python benchmarks/nccl_probe.py            # spawns 2 ranks
"""

import json
import os
import sys
import traceback

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def worker(rank):
    import torch.distributed as dist

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29671"
    device = torch.device("cuda:0")
    torch.cuda.set_device(device)
    dist.init_process_group("nccl", rank=rank, world_size=2, device_id=device)
    try:
        # grouped p2p exchange, both directions (the engine's C1/C2 shape)
        send = torch.full((64, 117), float(rank + 1), device=device)
        recv = torch.empty(64, 117, device=device)
        peer = 1 - rank
        ops = [
            dist.P2POp(dist.isend, send, peer),
            dist.P2POp(dist.irecv, recv, peer),
        ]
        for w in dist.batch_isend_irecv(ops):
            w.wait()
        torch.cuda.synchronize()
        assert float(recv.mean()) == float(peer + 1), float(recv.mean())
        dist.barrier()
        if rank == 0:
            print(json.dumps({"nccl_two_ranks_one_gpu": "ok"}), flush=True)
    finally:
        dist.destroy_process_group()


def engine_worker(rank):
    """2-rank engine run over real RCCL on one GPU; rank 0 prints params
    checksum for comparison with a 1-rank run."""
    import numpy as np
    import torch.distributed as dist

    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.data import make_synthetic_classification
    from gossipy_amd.engine import (
        BatchedGossipSimulator,
        DataArena,
        EngineConfig,
        LogRegSpec,
    )

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29673"
    device = torch.device("cuda:0")
    torch.cuda.set_device(device)
    dist.init_process_group("nccl", rank=rank, world_size=2, device_id=device)
    try:
        n = 40
        X, y = make_synthetic_classification((40 * n, 57, 2), seed=1)
        idx = np.random.default_rng(1).permutation(len(y))
        cut = int(0.9 * len(y))
        shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n)]
        half = n // 2
        lo = rank * half
        data = DataArena.from_shards(
            shards[lo : lo + half], device,
            global_eval=(X[idx[cut:]], y[idx[cut:]]),
        )
        cfg = EngineConfig(
            n_nodes=n, delta=10, protocol=AntiEntropyProtocol.PUSH_PULL,
            model_size=116, sampling_eval=0.25, seed=5,
        )
        sim = BatchedGossipSimulator(
            cfg, LogRegSpec(d_in=57, n_classes=2, lr=0.1), data, device=device
        )
        sim.init_nodes()
        sim.start(n_rounds=5)
        full = sim.gather_params()
        if rank == 0:
            print(json.dumps({
                "nccl_engine_2rank_checksum": float(full.double().sum().item())
            }), flush=True)
    finally:
        dist.destroy_process_group()


def single_rank_checksum():
    import numpy as np

    from gossipy_amd.core import AntiEntropyProtocol
    from gossipy_amd.data import make_synthetic_classification
    from gossipy_amd.engine import (
        BatchedGossipSimulator,
        DataArena,
        EngineConfig,
        LogRegSpec,
    )

    device = torch.device("cuda:0")
    n = 40
    X, y = make_synthetic_classification((40 * n, 57, 2), seed=1)
    idx = np.random.default_rng(1).permutation(len(y))
    cut = int(0.9 * len(y))
    shards = [(X[s], y[s]) for s in np.array_split(idx[:cut], n)]
    data = DataArena.from_shards(
        shards, device, global_eval=(X[idx[cut:]], y[idx[cut:]])
    )
    cfg = EngineConfig(
        n_nodes=n, delta=10, protocol=AntiEntropyProtocol.PUSH_PULL,
        model_size=116, sampling_eval=0.25, seed=5,
    )
    sim = BatchedGossipSimulator(
        cfg, LogRegSpec(d_in=57, n_classes=2, lr=0.1), data, device=device
    )
    sim.init_nodes()
    sim.start(n_rounds=5)
    print(json.dumps({
        "nccl_engine_1rank_checksum": float(
            sim.local_params().double().sum().item()
        )
    }), flush=True)


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "single":
        single_rank_checksum()
        sys.exit(0)
    import torch.multiprocessing as tmp

    for name, fn in (
        ("p2p", worker),
        ("engine", engine_worker),
    ):
        try:
            tmp.spawn(fn, nprocs=2, join=True)
            print(json.dumps({f"nccl_probe_{name}": "ok"}), flush=True)
        except Exception:
            traceback.print_exc()
            print(json.dumps({f"nccl_probe_{name}": "FAILED"}), flush=True)
            break
