"""Seed URL lists (reference parity: infomesh/crawler/seeds.py +
seeds/*.txt category lists)."""
from __future__ import annotations

from pathlib import Path

CATEGORIES = ("quickstart", "tech-docs", "academic", "encyclopedia",
              "search-strategy")


def seeds_dir() -> Path:
    return Path(__file__).resolve().parents[2] / "seeds"


def load_seeds(category: str = "quickstart",
               path: Path | None = None) -> list[str]:
    p = path or (seeds_dir() / f"{category}.txt")
    if not p.exists():
        return []
    urls = []
    for line in p.read_text(encoding="utf-8").splitlines():
        line = line.strip()
        if line and not line.startswith("#"):
            urls.append(line)
    return urls


def load_all_seeds() -> dict[str, list[str]]:
    return {c: load_seeds(c) for c in CATEGORIES}
