"""Distributed launch helpers (single node over xGMI; rendezvous on
127.0.0.1 — reference: utils/parallel_training_utils.py:4-9,
strategy.py:288-289)."""

import os
import socket

import torch.distributed as dist


def get_free_tcp_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("", 0))
        return s.getsockname()[1]


def setup_master_env(port=None):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ["MASTER_PORT"] = str(port or get_free_tcp_port())


def init_process_group_from_env(backend=None, rank=None, world_size=None):
    """Init torch.distributed; backend 'nccl' (= RCCL) on GPU, 'gloo' on CPU."""
    import torch
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend, rank=rank, world_size=world_size)
    return backend
