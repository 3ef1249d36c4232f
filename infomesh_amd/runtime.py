"""Runtime coordination: PID file, startup lock, status heartbeat.

Reference parity: infomesh/runtime.py (PID file + /proc cmdline
validation, StartupLock flock, runtime_status.json heartbeat with 30 s
staleness, SIGTERM graceful stop).
"""
from __future__ import annotations

import fcntl
import json
import os
import signal
import time
from pathlib import Path

from .errors import InfoMeshError

HEARTBEAT_INTERVAL_S = 10.0
STALE_AFTER_S = 30.0


def _cmdline_of(pid: int) -> str:
    try:
        return Path(f"/proc/{pid}/cmdline").read_bytes()\
            .replace(b"\x00", b" ").decode(errors="replace")
    except OSError:
        return ""


class PidFile:
    """PID file with liveness + cmdline validation against PID reuse."""

    def __init__(self, data_dir: Path, name: str = "infomesh.pid",
                 marker: str = "infomesh"):
        self.path = data_dir / name
        self.marker = marker

    def read_running_pid(self) -> int | None:
        if not self.path.exists():
            return None
        try:
            pid = int(self.path.read_text().strip())
        except (ValueError, OSError):
            return None
        if pid <= 0:
            return None
        try:
            os.kill(pid, 0)
        except ProcessLookupError:
            return None
        except PermissionError:
            pass
        cmd = _cmdline_of(pid)
        if cmd and self.marker not in cmd:
            return None  # PID reused by an unrelated process
        return pid

    def acquire(self) -> None:
        pid = self.read_running_pid()
        if pid is not None and pid != os.getpid():
            raise InfoMeshError("RT001", f"pid {pid}")
        self.path.parent.mkdir(parents=True, exist_ok=True)
        tmp = self.path.with_suffix(".tmp")
        tmp.write_text(str(os.getpid()))
        tmp.replace(self.path)

    def release(self) -> None:
        try:
            if self.read_running_pid() == os.getpid():
                self.path.unlink(missing_ok=True)
        except OSError:
            pass


class StartupLock:
    """Cross-process flock held for the whole process lifetime."""

    def __init__(self, data_dir: Path, name: str = "startup.lock"):
        self.path = data_dir / name
        self._fh = None

    def acquire(self, blocking: bool = False) -> bool:
        self.path.parent.mkdir(parents=True, exist_ok=True)
        self._fh = open(self.path, "w")
        try:
            flags = fcntl.LOCK_EX | (0 if blocking else fcntl.LOCK_NB)
            fcntl.flock(self._fh, flags)
            self._fh.write(str(os.getpid()))
            self._fh.flush()
            return True
        except BlockingIOError:
            self._fh.close()
            self._fh = None
            return False

    def release(self) -> None:
        if self._fh is not None:
            try:
                fcntl.flock(self._fh, fcntl.LOCK_UN)
            finally:
                self._fh.close()
                self._fh = None

    def __enter__(self):
        if not self.acquire():
            raise InfoMeshError("RT001", "startup lock held")
        return self

    def __exit__(self, *exc):
        self.release()
        return False


class RuntimeStatus:
    """Atomic-write heartbeat file readable by dashboard/API processes."""

    def __init__(self, data_dir: Path, name: str = "runtime_status.json"):
        self.path = data_dir / name

    def write(self, state: str = "running", **extra) -> None:
        payload = {"state": state, "pid": os.getpid(),
                   "ts": time.time(), **extra}
        self.path.parent.mkdir(parents=True, exist_ok=True)
        tmp = self.path.with_suffix(".tmp")
        tmp.write_text(json.dumps(payload))
        tmp.replace(self.path)

    def read(self) -> dict:
        try:
            data = json.loads(self.path.read_text())
        except (OSError, json.JSONDecodeError):
            return {"state": "stopped", "stale": True}
        if time.time() - data.get("ts", 0) > STALE_AFTER_S:
            data["state"] = "stopped"
            data["stale"] = True
        return data


class GracefulShutdown:
    """SIGTERM/SIGINT orchestration (reference: shutdown.py:19)."""

    def __init__(self):
        self.requested = False
        self._callbacks = []

    def install(self) -> None:
        for sig in (signal.SIGTERM, signal.SIGINT):
            signal.signal(sig, self._handler)

    def _handler(self, signum, frame) -> None:
        self.requested = True
        for cb in self._callbacks:
            try:
                cb()
            except Exception:
                pass

    def on_shutdown(self, cb) -> None:
        self._callbacks.append(cb)
