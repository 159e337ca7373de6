"""Runtime configuration (replaces the reference's compile-time defines,
env vars and global statics — SURVEY.md §5.6; cf. reference main.cpp:64-73,
Makefile:27-40).

One dataclass, constructed from kwargs / CLI / env — role selection is a
runtime flag, hyperparameters are per-trainer fields with the same defaults
the reference hard-codes.
"""

from __future__ import annotations

import dataclasses
import os
from dataclasses import dataclass, field
from typing import Optional


def _env_int(name: str, default: int) -> int:
    v = os.environ.get(name)
    return int(v) if v is not None else default


def _env_str(name: str, default: str) -> str:
    return os.environ.get(name, default)


@dataclass
class TrainConfig:
    """Hyperparameters shared by the trainers.

    Mirrors the reference's GradientUpdater globals (main.cpp:64-73):
    __global_minibatch_size / learning_rate / ema_rate / sparse_rate /
    lambdaL2 / lambdaL1.
    """

    minibatch_size: int = 64
    learning_rate: float = 0.1
    ema_rate: float = 0.9
    sparse_rate: float = 0.0  # dropout rate on FC outputs
    lambda_l2: float = 1e-3
    lambda_l1: float = 1e-4
    epochs: int = 5
    optimizer: str = "adagrad"  # sgd|adagrad|rmsprop|ftrl|adadelta|adam
    dtype: str = "fp32"  # compute dtype for dense paths: fp32|bf16
    seed: int = 1234


@dataclass
class ClusterConfig:
    """Cluster shape (replaces LightCTR_PS_NUM / LightCTR_WORKER_NUM /
    LightCTR_MASTER_ADDR env config, reference master.h:23-24, network.h:36-38).

    On MI355X this is one process per GPU in a single node over RCCL;
    rendezvous via torch.distributed (MASTER_ADDR/MASTER_PORT env).
    """

    world_size: int = field(default_factory=lambda: _env_int("WORLD_SIZE", 1))
    rank: int = field(default_factory=lambda: _env_int("RANK", 0))
    local_rank: int = field(default_factory=lambda: _env_int("LOCAL_RANK", 0))
    master_addr: str = field(default_factory=lambda: _env_str("MASTER_ADDR", "127.0.0.1"))
    master_port: int = field(default_factory=lambda: _env_int("MASTER_PORT", 29500))
    backend: Optional[str] = None  # None => nccl(RCCL) if cuda available else gloo
    # PS mode: number of ranks acting as parameter shards (rest are workers).
    # 0 => pure data-parallel (ring) mode.
    ps_shards: int = 0

    def resolved_backend(self) -> str:
        if self.backend:
            return self.backend
        import torch

        return "nccl" if torch.cuda.is_available() else "gloo"


def asdict(cfg) -> dict:
    return dataclasses.asdict(cfg)
