"""Process-group fabric helpers + rank-fault tolerance.

Design note: xGMI is point-to-point (7 links/GPU), so
for the tiny per-query payloads here latency dominates — we use one
direct broadcast + one all-gather per batch (no rings, no trees), and
batch queries so each collective carries a full batch.

Fault model (reference parity: infomesh/search/query.py:471-490 serves
local-only when peers are absent): crash-stop rank failures. Every rank
heartbeats into a TCPStore sidecar; one exclusion subgroup per rank is
pre-created at startup (subgroup communicators are independent of the
default group's, so they stay healthy after a default-group collective
times out on a dead rank). On a collective failure the survivors read
the heartbeats, agree on the dead rank, and switch to its exclusion
subgroup — single-rank failures keep serving on W-1 shards (flagged
degraded); anything worse degrades to local-only.
"""
from __future__ import annotations

import datetime
import logging
import os
import threading
import time

import torch
import torch.distributed as dist

log = logging.getLogger("infomesh.fabric")

HB_KEY = "imhb_{rank}"
LOCAL = "local"     # sentinel group: no collectives, serve own shard


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


class Fabric:
    """Thin wrapper over a torch.distributed process group."""

    def __init__(self, backend: str | None = None,
                 timeout_s: float = 300.0, heartbeat_s: float = 1.0,
                 fault_tolerant: bool = True):
        self.rank = env_rank()
        self.world = env_world()
        self.timeout_s = timeout_s
        self.heartbeat_s = heartbeat_s
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
        # --- fault tolerance state ---
        self.group = None                       # None = default group
        self.active_ranks = list(range(self.world))
        self.degraded = False
        self._excl: dict[int, object] = {}
        self._hb_store = None
        self._ctl_store = None
        self._hb_stop = threading.Event()
        if self.initialized and fault_tolerant:
            self._init_fault_tolerance()

    def _init_fault_tolerance(self) -> None:
        # one exclusion subgroup per possibly-dead rank; ALL ranks must
        # participate in creation, so it happens here at startup while
        # everyone is alive
        for r in range(self.world):
            ranks = [x for x in range(self.world) if x != r]
            self._excl[r] = dist.new_group(
                ranks=ranks,
                timeout=datetime.timedelta(seconds=self.timeout_s))
        # TWO sidecar TCPStore clients (own connections — never contend
        # with the process group's store use): one owned by the
        # heartbeat thread, one by the main thread's health checks.
        # They must be separate: a long resync wait() on a shared
        # client would starve the beats and make THIS rank look dead.
        try:
            self._hb_store = self._new_store_client()
            self._ctl_store = self._new_store_client()
            self._beat()
            t = threading.Thread(target=self._hb_loop,
                                 name="infomesh-heartbeat", daemon=True)
            t.start()
        except Exception as e:   # heartbeats are best-effort
            log.warning("heartbeat store unavailable: %s", e)
            self._hb_store = None
            self._ctl_store = None
        dist.barrier()           # everyone wrote its first heartbeat

    @staticmethod
    def _new_store_client():
        return dist.TCPStore(
            os.environ.get("MASTER_ADDR", "127.0.0.1"),
            int(os.environ.get("MASTER_PORT", "29500")),
            is_master=False,
            timeout=datetime.timedelta(seconds=10))

    def _beat(self) -> None:
        self._hb_store.set(HB_KEY.format(rank=self.rank),
                           str(time.time()))

    def _hb_loop(self) -> None:
        while not self._hb_stop.wait(self.heartbeat_s):
            try:
                self._beat()
            except Exception:
                return

    # ------------------------------------------------- fault handling
    def dead_ranks(self, stale_s: float = 5.0) -> list[int]:
        """Ranks whose heartbeat is stale (single node — one clock)."""
        if not self.initialized or self._ctl_store is None:
            return []
        dead = []
        now = time.time()
        for r in self.active_ranks:
            if r == self.rank:
                continue
            try:
                ts = float(self._ctl_store.get(
                    HB_KEY.format(rank=r)).decode())
            except Exception:
                dead.append(r)
                continue
            if now - ts > stale_s:
                dead.append(r)
        return dead

    def resync(self, wait_s: float | None = None) -> bool:
        """Store-based rendezvous of the ACTIVE ranks after a degrade —
        deliberately collective-free: a timed-out op CLOSES its gloo
        pairs (and may poison an NCCL comm), so the fresh subgroup must
        not be touched until every survivor has drained its own failure
        and arrived here. Returns False if some active rank never
        showed up (caller should degrade further)."""
        if self._ctl_store is None or not self.collective_ok:
            return False
        wait_s = wait_s if wait_s is not None else self.timeout_s + 5.0
        tag = ".".join(map(str, self.active_ranks))
        self._ctl_store.set(f"imrec_{tag}_{self.rank}", "1")
        keys = [f"imrec_{tag}_{r}" for r in self.active_ranks]
        try:
            self._ctl_store.wait(
                keys, datetime.timedelta(seconds=wait_s))
            return True
        except Exception:
            return False

    def degrade(self, dead: list[int]) -> None:
        """Shrink to the pre-created exclusion subgroup (single dead
        rank) or to local-only (anything worse)."""
        self.degraded = True
        if len(dead) == 1 and dead[0] in self._excl and self.group is None:
            self.group = self._excl[dead[0]]
            self.active_ranks = [x for x in self.active_ranks
                                 if x != dead[0]]
            log.warning("rank %d dead -> serving degraded on ranks %s",
                        dead[0], self.active_ranks)
        else:
            self.group = LOCAL
            self.active_ranks = [self.rank]
            log.warning("multiple/unknown rank failures %s -> local-only",
                        dead)

    @property
    def collective_ok(self) -> bool:
        return self.initialized and self.group is not LOCAL

    @property
    def effective_world(self) -> int:
        return len(self.active_ranks)

    # ------------------------------------------------------ collectives
    @property
    def initialized(self) -> bool:
        return self.world > 1 and dist.is_initialized()

    @property
    def device(self) -> torch.device:
        if torch.cuda.is_available():
            return torch.device("cuda", env_local_rank())
        return torch.device("cpu")

    def barrier(self) -> None:
        if self.collective_ok:
            dist.barrier(group=self.group)

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.collective_ok:
            dist.broadcast(t, src=src, group=self.group)
        return t

    def all_gather(self, t: torch.Tensor) -> torch.Tensor:
        """Gather equal-shaped tensors from the ACTIVE ranks ->
        [effective_world, *shape]."""
        if not self.collective_ok:
            return t.unsqueeze(0)
        out = [torch.empty_like(t) for _ in self.active_ranks]
        dist.all_gather(out, t.contiguous(), group=self.group)
        return torch.stack(out, dim=0)

    def all_reduce_max(self, value: float) -> float:
        if not self.collective_ok:
            return value
        t = torch.tensor([value], dtype=torch.float64,
                         device=self.device if self.backend == "nccl"
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=self.group)
        return float(t.item())

    def destroy(self) -> None:
        self._hb_stop.set()
        if dist.is_initialized():
            dist.destroy_process_group()
