"""Resource governor: CPU/memory/HBM sampling -> degradation ladder.

Reference parity: infomesh/resources/governor.py (DegradeLevel 0-4:
normal / throttle crawl / pause crawl / disable remote search /
read-only; throttle factor, RSS limit before the OOM killer).
MI355X addition: HBM headroom on the local GPU feeds the same ladder.
"""
from __future__ import annotations

import enum
import os
import time
from dataclasses import dataclass, field


class DegradeLevel(enum.IntEnum):
    NORMAL = 0
    THROTTLE_CRAWL = 1
    PAUSE_CRAWL = 2
    LOCAL_ONLY = 3
    READ_ONLY = 4


def _read_meminfo() -> tuple[float, float]:
    """(available_gb, total_gb) from /proc/meminfo."""
    avail = total = 0.0
    try:
        for line in open("/proc/meminfo"):
            if line.startswith("MemAvailable:"):
                avail = int(line.split()[1]) / 1e6
            elif line.startswith("MemTotal:"):
                total = int(line.split()[1]) / 1e6
    except OSError:
        pass
    return avail, total


def _rss_gb() -> float:
    try:
        with open(f"/proc/{os.getpid()}/statm") as f:
            pages = int(f.read().split()[1])
        return pages * os.sysconf("SC_PAGE_SIZE") / 1e9
    except (OSError, ValueError):
        return 0.0


def _load_per_cpu() -> float:
    try:
        return os.getloadavg()[0] / max(1, os.cpu_count() or 1)
    except OSError:
        return 0.0


def _hbm_headroom_frac() -> float:
    """Free fraction of HBM on the current device (1.0 = all free)."""
    try:
        import torch
        if not torch.cuda.is_available():
            return 1.0
        free, total = torch.cuda.mem_get_info()
        return free / max(1, total)
    except Exception:
        return 1.0


@dataclass
class ResourceSample:
    load_per_cpu: float
    mem_available_gb: float
    mem_total_gb: float
    rss_gb: float
    hbm_free_frac: float
    ts: float = field(default_factory=time.time)


@dataclass
class ResourceGovernor:
    max_rss_gb: float = 32.0
    min_mem_available_gb: float = 2.0
    max_load_per_cpu: float = 2.0
    min_hbm_free_frac: float = 0.03
    interval_s: float = 5.0
    _last: ResourceSample | None = None
    _last_ts: float = 0.0
    level: DegradeLevel = DegradeLevel.NORMAL

    def sample(self, force: bool = False) -> ResourceSample:
        now = time.time()
        if not force and self._last is not None and \
                now - self._last_ts < self.interval_s:
            return self._last
        avail, total = _read_meminfo()
        s = ResourceSample(_load_per_cpu(), avail, total, _rss_gb(),
                           _hbm_headroom_frac())
        self._last, self._last_ts = s, now
        self.level = self._level_for(s)
        return s

    def _level_for(self, s: ResourceSample) -> DegradeLevel:
        if s.mem_available_gb < self.min_mem_available_gb / 2 or \
                s.rss_gb > self.max_rss_gb:
            return DegradeLevel.READ_ONLY
        if s.hbm_free_frac < self.min_hbm_free_frac:
            return DegradeLevel.LOCAL_ONLY
        if s.mem_available_gb < self.min_mem_available_gb:
            return DegradeLevel.PAUSE_CRAWL
        if s.load_per_cpu > self.max_load_per_cpu:
            return DegradeLevel.THROTTLE_CRAWL
        return DegradeLevel.NORMAL

    def throttle_factor(self) -> float:
        """Crawl-delay multiplier (1.0 = no throttle)."""
        self.sample()
        return {DegradeLevel.NORMAL: 1.0,
                DegradeLevel.THROTTLE_CRAWL: 3.0,
                DegradeLevel.PAUSE_CRAWL: float("inf"),
                DegradeLevel.LOCAL_ONLY: float("inf"),
                DegradeLevel.READ_ONLY: float("inf")}[self.level]

    def crawl_allowed(self) -> bool:
        self.sample()
        return self.level < DegradeLevel.PAUSE_CRAWL

    def writes_allowed(self) -> bool:
        self.sample()
        return self.level < DegradeLevel.READ_ONLY


# Resource profiles (reference: resources/profiles.py)
PROFILES = {
    "minimal": {"max_rss_gb": 4.0, "max_load_per_cpu": 1.0},
    "balanced": {"max_rss_gb": 16.0, "max_load_per_cpu": 2.0},
    "contributor": {"max_rss_gb": 64.0, "max_load_per_cpu": 4.0},
    "dedicated": {"max_rss_gb": 512.0, "max_load_per_cpu": 16.0},
}


def governor_for_profile(profile: str = "balanced") -> ResourceGovernor:
    kw = PROFILES.get(profile, PROFILES["balanced"])
    return ResourceGovernor(**kw)


def run_preflight_checks(data_dir, min_disk_mb: int = 200) -> list[str]:
    """Disk/GPU checks before start (reference: preflight.py:71-192;
    the outbound-connectivity check is skipped in offline deployments).
    Returns a list of failure strings (empty = OK)."""
    import shutil
    problems = []
    try:
        data_dir.mkdir(parents=True, exist_ok=True)
        free_mb = shutil.disk_usage(str(data_dir)).free / 1e6
        if free_mb < min_disk_mb:
            problems.append(f"only {free_mb:.0f} MB disk free "
                            f"(need {min_disk_mb})")
    except OSError as e:
        problems.append(f"data dir not writable: {e}")
    try:
        import torch
        if torch.cuda.is_available():
            from ..ops import _ext
            if not _ext.available():
                problems.append("GPU present but HIP extension not built "
                                "(python -m infomesh_amd.ops._build)")
    except Exception as e:
        problems.append(f"torch probe failed: {e}")
    return problems
