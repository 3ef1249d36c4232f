#!/usr/bin/env python3
"""Driver-contract self-check (run on CPU; fast).

Validates the pieces the round driver depends on:
  * bench.py emits the exact JSON contract keys (CPU sanity mode),
  * __graft_entry__ exposes build() and smoke(),
  * every GPU test is marked and the marker is registered,
  * the HIP extension source set compiles (hipcc cross-compile smoke
    is covered by build(); here we only check the sources exist).
"""
from __future__ import annotations

import json
import pathlib
import subprocess
import sys

ROOT = pathlib.Path(__file__).resolve().parents[1]
REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling",
                 "vs_baseline", "dtype", "data", "config"}
REQUIRED_CONFIG = {"model", "global_batch", "seq_len", "parallelism"}


def check_bench() -> None:
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--docs-per-gpu", "3000", "--batch", "8"],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    missing = REQUIRED_KEYS - set(d)
    assert not missing, f"bench JSON missing {missing}"
    missing_c = REQUIRED_CONFIG - set(d["config"])
    assert not missing_c, f"bench config missing {missing_c}"
    assert isinstance(d["value"], (int, float)) and d["value"] > 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    print(f"bench contract OK ({d['value']} {d['unit']} CPU-sanity)")


def check_graft_entry() -> None:
    sys.path.insert(0, str(ROOT))
    import __graft_entry__ as g
    assert callable(g.build) and callable(g.smoke)
    print("graft entry OK (build/smoke present)")


def check_markers() -> None:
    conftest = (ROOT / "tests" / "conftest.py").read_text()
    assert "gpu" in conftest, "gpu marker must be registered"
    out = subprocess.run(
        [sys.executable, "-m", "pytest", "tests", "-q", "-m", "gpu",
         "--collect-only"], cwd=ROOT, capture_output=True, text=True,
        timeout=300)
    assert out.returncode in (0, 5)
    print("markers OK")


def check_sources() -> None:
    hips = list((ROOT / "infomesh_amd" / "ops" / "csrc").glob("*.hip"))
    assert len(hips) >= 10, f"expected HIP sources, found {len(hips)}"
    print(f"sources OK ({len(hips)} .hip files)")


if __name__ == "__main__":
    check_sources()
    check_graft_entry()
    check_markers()
    check_bench()
    print("driver contract: ALL OK")
