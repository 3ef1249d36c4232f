"""Process-group fabric helpers.

Design note: xGMI is point-to-point (7 links/GPU), so
for the tiny per-query payloads here latency dominates — we use one
direct broadcast + one all-gather per batch (no rings, no trees), and
batch queries so each collective carries a full batch.
"""
from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


class Fabric:
    """Thin wrapper over a torch.distributed process group."""

    def __init__(self, backend: str | None = None,
                 timeout_s: float = 300.0):
        self.rank = env_rank()
        self.world = env_world()
        # Bind this process to its GPU BEFORE the process group exists:
        # otherwise every rank's first CUDA touch lands on device 0 and
        # RCCL communicators bind to the wrong device.
        use_cuda = torch.cuda.is_available()
        if use_cuda and self.world > 1:
            torch.cuda.set_device(env_local_rank())
        if self.world > 1 and not dist.is_initialized():
            if backend is None:
                backend = "nccl" if use_cuda else "gloo"
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29500")
            kwargs = {}
            if backend == "nccl" and use_cuda:
                kwargs["device_id"] = torch.device("cuda",
                                                   env_local_rank())
            dist.init_process_group(
                backend=backend, rank=self.rank, world_size=self.world,
                timeout=datetime.timedelta(seconds=timeout_s), **kwargs)
        self.backend = dist.get_backend() if dist.is_initialized() else "none"

    @property
    def initialized(self) -> bool:
        return self.world > 1 and dist.is_initialized()

    @property
    def device(self) -> torch.device:
        if torch.cuda.is_available():
            return torch.device("cuda", env_local_rank())
        return torch.device("cpu")

    def barrier(self) -> None:
        if self.initialized:
            dist.barrier()

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.initialized:
            dist.broadcast(t, src=src)
        return t

    def all_gather(self, t: torch.Tensor) -> torch.Tensor:
        """Gather equal-shaped tensors from every rank -> [world, *shape]."""
        if not self.initialized:
            return t.unsqueeze(0)
        out = [torch.empty_like(t) for _ in range(self.world)]
        dist.all_gather(out, t.contiguous())
        return torch.stack(out, dim=0)

    def all_reduce_max(self, value: float) -> float:
        if not self.initialized:
            return value
        t = torch.tensor([value], dtype=torch.float64,
                         device=self.device if self.backend == "nccl"
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())

    def destroy(self) -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
