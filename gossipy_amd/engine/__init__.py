"""The batched MI355X engine.

Executes the gossip simulation as node-batched kernels over packed HBM
arenas instead of per-node Python dispatch: the event schedule is derived
deterministically on every rank (:mod:`.schedule`), per-tick work runs as a
handful of batched kernel launches (:mod:`.backend`), and cross-GPU model
traffic travels as grouped RCCL point-to-point transfers over xGMI
(:mod:`.runner`). See SURVEY.md §2.4/§2.5 for the site-by-site mapping from
the reference.
"""

from .arena import DataArena, NodeStateArena, SlotPool
from .backend import HIPBackend, TorchBackend, make_backend
from .models import (
    AdaLineSpec,
    KMeansSpec,
    LogRegSpec,
    MFSpec,
    MLPSpec,
    PegasosSpec,
    TorchModuleSpec,
)
from .rng import Purpose, RandomTape
from .runner import (
    BatchedAll2AllGossipSimulator,
    BatchedCacheNeighGossipSimulator,
    BatchedGossipSimulator,
    BatchedPENSGossipSimulator,
    BatchedTokenizedGossipSimulator,
    RoundTimer,
)
from .schedule import (
    EngineConfig,
    RoundSchedule,
    Scheduler,
    TickPhase,
    TokenizedScheduler,
    make_scheduler,
)

__all__ = [
    "BatchedGossipSimulator",
    "BatchedTokenizedGossipSimulator",
    "BatchedAll2AllGossipSimulator",
    "BatchedCacheNeighGossipSimulator",
    "BatchedPENSGossipSimulator",
    "TokenizedScheduler",
    "EngineConfig",
    "Scheduler",
    "RoundSchedule",
    "TickPhase",
    "RandomTape",
    "Purpose",
    "DataArena",
    "NodeStateArena",
    "SlotPool",
    "TorchBackend",
    "HIPBackend",
    "make_backend",
    "AdaLineSpec",
    "PegasosSpec",
    "LogRegSpec",
    "MLPSpec",
    "MFSpec",
    "KMeansSpec",
    "TorchModuleSpec",
    "RoundTimer",
]
