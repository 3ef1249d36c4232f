"""`doctor` diagnostics (reference parity: infomesh/diagnostics.py —
10 environment/health checks + bench hooks)."""
from __future__ import annotations

import shutil
import sqlite3
from typing import Any


def _check(name: str, fn) -> dict[str, Any]:
    try:
        ok, detail = fn()
    except Exception as e:
        ok, detail = False, f"{type(e).__name__}: {e}"
    return {"name": name, "ok": bool(ok), "detail": str(detail)}


def run_doctor() -> dict[str, Any]:
    from ..config import load_config
    cfg = load_config()

    def c_python():
        import sys
        v = sys.version_info
        return v >= (3, 10), f"python {v.major}.{v.minor}"

    def c_torch():
        import torch
        return True, f"torch {torch.__version__}"

    def c_gpu():
        import torch
        if not torch.cuda.is_available():
            return True, "no GPU (CPU mode)"
        return True, f"{torch.cuda.device_count()}× {torch.cuda.get_device_name(0)}"

    def c_ext():
        import torch
        from ..ops import _ext
        if not torch.cuda.is_available():
            return True, "skipped (no GPU)"
        return _ext.available(), \
            "loaded" if _ext.available() else "NOT BUILT — run ops._build"

    def c_hipcc():
        return shutil.which("hipcc") is not None or \
            shutil.which("/opt/rocm/bin/hipcc") is not None, \
            shutil.which("hipcc") or "/opt/rocm/bin/hipcc?"

    def c_fts5():
        conn = sqlite3.connect(":memory:")
        conn.execute("CREATE VIRTUAL TABLE t USING fts5(x)")
        return True, "FTS5 available"

    def c_zstd():
        from .. import compression
        data = compression.decompress(compression.compress(b"x" * 100))
        return data == b"x" * 100, "libzstd roundtrip"

    def c_disk():
        free = shutil.disk_usage(str(cfg.data_dir.parent
                                     if not cfg.data_dir.exists()
                                     else cfg.data_dir)).free / 1e6
        return free > 200, f"{free:.0f} MB free"

    def c_datadir():
        cfg.data_dir.mkdir(parents=True, exist_ok=True)
        probe = cfg.data_dir / ".probe"
        probe.write_text("x")
        probe.unlink()
        return True, str(cfg.data_dir)

    def c_keys():
        from ..trust.keys import ensure_keys
        kp = ensure_keys(cfg.data_dir)
        return True, f"node {kp.node_id[:12]}"

    def c_fabric():
        import torch.distributed as dist
        backends = [b for b in ("nccl", "gloo")
                    if getattr(dist, f"is_{b}_available")()]
        return bool(backends), "+".join(backends)

    checks = [
        _check("python", c_python),
        _check("torch", c_torch),
        _check("gpu", c_gpu),
        _check("hip extension", c_ext),
        _check("hipcc", c_hipcc),
        _check("sqlite fts5", c_fts5),
        _check("zstd", c_zstd),
        _check("disk space", c_disk),
        _check("data dir", c_datadir),
        _check("node keys", c_keys),
        _check("dist backends", c_fabric),
    ]
    return {"ok": all(c["ok"] for c in checks), "checks": checks}
