"""Configuration system.

Reference parity: infomesh/config.py (frozen dataclasses per section,
TOML at ~/.infomesh/config.toml, env overrides INFOMESH_{SECTION}_{KEY},
range clamping + enum whitelists, node roles, save only non-defaults).

MI355X additions: a [gpu] section (shard count, HBM budget, dtype, kernel
toggles) replacing the reference's [p2p] network knobs for the intra-node
fabric.
"""
from __future__ import annotations

import dataclasses
import logging
import os
import re
from dataclasses import dataclass, field, fields
from pathlib import Path
from typing import Any

log = logging.getLogger("infomesh.config")

NODE_ROLES = ("full", "crawler", "search")
FTS_TOKENIZERS = ("unicode61", "ascii", "porter", "trigram")
COMPUTE_DTYPES = ("bf16", "fp16", "fp32", "fp8")


def default_data_dir() -> Path:
    env = os.environ.get("INFOMESH_DATA_DIR")
    if env:
        return Path(env)
    return Path.home() / ".infomesh"


@dataclass(frozen=True)
class NodeConfig:
    role: str = "full"                 # full | crawler | search
    data_dir: str = ""                 # empty → ~/.infomesh
    name: str = ""
    plugins: str = ""                  # comma-separated dotted paths


@dataclass(frozen=True)
class CrawlConfig:
    max_concurrent: int = 5            # reference default (config.py:61-66)
    politeness_delay_s: float = 1.0
    max_urls_per_hour: int = 60
    max_depth: int = 3
    max_response_bytes: int = 10_000_000
    retries: int = 2
    timeout_s: float = 20.0
    respect_robots: bool = True
    user_agent: str = "infomesh-amd/0.1 (+https://github.com/infomesh)"


@dataclass(frozen=True)
class IndexConfig:
    fts_tokenizer: str = "unicode61"
    max_text_chars: int = 500_000
    embed_max_chars: int = 2000        # reference: vector_store.py:144-157
    snapshot_compression_level: int = 12


@dataclass(frozen=True)
class SearchConfig:
    max_results: int = 10
    max_results_per_shard: int = 20    # mirror of MAX_RESULTS_PER_PEER (routing.py:48)
    cache_entries: int = 1000          # reference: mcp/server.py:120-126
    cache_ttl_s: float = 300.0
    rrf_k: int = 60                    # reference: search/merge.py
    rerank_top_n: int = 100
    rerank_keep: int = 10
    hybrid: bool = True
    # dynamic micro-batching between entry points and the GPU plane
    batch_max: int = 128               # matches the benched batch size
    batch_wait_ms: float = 1.5         # max collect latency per request


@dataclass(frozen=True)
class GpuConfig:
    """MI355X fabric settings (replaces the reference's [p2p] section
    for the intra-node path)."""
    n_shards: int = 0                  # 0 → world size at runtime
    hbm_budget_gb: float = 260.0       # of 288 GB HBM3E per GPU
    dtype: str = "bf16"
    topk_per_shard: int = 100
    require_extension: bool = True     # GPU present but ext missing → hard error


@dataclass(frozen=True)
class CreditsConfig:
    enabled: bool = True
    search_cost: float = 0.100         # tier-1 (ledger.py:12-15)
    crawl_reward: float = 1.0
    query_reward: float = 0.5
    grace_hours: float = 72.0


@dataclass(frozen=True)
class TrustConfig:
    enabled: bool = True
    audits_per_hour: float = 1.0       # trust/audit.py:28
    auditors: int = 3
    isolation_failures: int = 3


@dataclass(frozen=True)
class ApiConfig:
    host: str = "127.0.0.1"
    port: int = 8080
    api_key: str = ""
    rate_limit_per_min: int = 120
    # OTLP/HTTP collector base url ("http://host:4318"); empty = off
    otlp_endpoint: str = ""


@dataclass(frozen=True)
class SummarizerConfig:
    enabled: bool = True
    max_new_tokens: int = 128
    max_context_tokens: int = 2048
    temperature: float = 0.0


@dataclass(frozen=True)
class Config:
    node: NodeConfig = field(default_factory=NodeConfig)
    crawl: CrawlConfig = field(default_factory=CrawlConfig)
    index: IndexConfig = field(default_factory=IndexConfig)
    search: SearchConfig = field(default_factory=SearchConfig)
    gpu: GpuConfig = field(default_factory=GpuConfig)
    credits: CreditsConfig = field(default_factory=CreditsConfig)
    trust: TrustConfig = field(default_factory=TrustConfig)
    api: ApiConfig = field(default_factory=ApiConfig)
    summarizer: SummarizerConfig = field(default_factory=SummarizerConfig)

    @property
    def data_dir(self) -> Path:
        return Path(self.node.data_dir) if self.node.data_dir else default_data_dir()


# ---------------------------------------------------------------- clamping

_CLAMPS: dict[tuple[str, str], tuple[float, float]] = {
    ("crawl", "max_concurrent"): (1, 64),
    ("crawl", "politeness_delay_s"): (0.0, 60.0),
    ("crawl", "max_urls_per_hour"): (1, 100_000),
    ("crawl", "max_depth"): (0, 10),
    ("crawl", "retries"): (0, 5),
    ("search", "max_results"): (1, 100),
    ("search", "max_results_per_shard"): (1, 1000),
    ("search", "cache_entries"): (0, 100_000),
    ("search", "rrf_k"): (1, 1000),
    ("gpu", "n_shards"): (0, 8),
    ("gpu", "hbm_budget_gb"): (1.0, 288.0),
    ("gpu", "topk_per_shard"): (1, 4096),
    ("api", "port"): (1, 65535),
}

_ENUMS: dict[tuple[str, str], tuple[str, ...]] = {
    ("node", "role"): NODE_ROLES,
    ("index", "fts_tokenizer"): FTS_TOKENIZERS,
    ("gpu", "dtype"): COMPUTE_DTYPES,
}


def _coerce(current: Any, raw: Any) -> Any:
    """Coerce a raw (string/TOML) value to the type of the default."""
    if isinstance(current, bool):
        if isinstance(raw, bool):
            return raw
        return str(raw).strip().lower() in ("1", "true", "yes", "on")
    if isinstance(current, int) and not isinstance(current, bool):
        return int(float(raw))
    if isinstance(current, float):
        return float(raw)
    return str(raw)


def _apply(section_name: str, section: Any, key: str, raw: Any) -> Any:
    if not hasattr(section, key):
        log.warning("config: unknown key [%s] %s ignored", section_name, key)
        return section
    cur = getattr(section, key)
    try:
        val = _coerce(cur, raw)
    except (TypeError, ValueError):
        log.warning("config: bad value for [%s] %s=%r ignored", section_name, key, raw)
        return section
    clamp = _CLAMPS.get((section_name, key))
    if clamp is not None and isinstance(val, (int, float)):
        lo, hi = clamp
        if val < lo or val > hi:
            clamped = min(max(val, lo), hi)
            log.warning("config: [%s] %s=%r clamped to %r", section_name, key, val, clamped)
            val = type(val)(clamped)
    enum = _ENUMS.get((section_name, key))
    if enum is not None and val not in enum:
        log.warning("config: [%s] %s=%r not in %s — keeping %r",
                    section_name, key, val, enum, cur)
        return section
    return dataclasses.replace(section, **{key: val})


# ---------------------------------------------------------------- TOML I/O
# Minimal TOML subset parser (tomllib is py3.11+; this image is 3.10):
# [section] headers, key = "string" | number | true/false lines, # comments.

_SECTION_RE = re.compile(r"^\[([A-Za-z0-9_]+)\]\s*$")
_KV_RE = re.compile(r"^([A-Za-z0-9_]+)\s*=\s*(.+?)\s*$")


def _parse_toml_value(text: str) -> Any:
    text = text.strip()
    if text.startswith('"') and text.endswith('"') and len(text) >= 2:
        return text[1:-1].encode().decode("unicode_escape")
    if text.startswith("'") and text.endswith("'") and len(text) >= 2:
        return text[1:-1]
    low = text.lower()
    if low == "true":
        return True
    if low == "false":
        return False
    try:
        if re.fullmatch(r"[+-]?\d+", text):
            return int(text)
        return float(text)
    except ValueError:
        return text


def parse_toml(text: str) -> dict[str, dict[str, Any]]:
    out: dict[str, dict[str, Any]] = {}
    section = ""
    for line in text.splitlines():
        line = line.split("#", 1)[0].strip() if not line.strip().startswith('"') else line.strip()
        if not line:
            continue
        m = _SECTION_RE.match(line)
        if m:
            section = m.group(1)
            out.setdefault(section, {})
            continue
        m = _KV_RE.match(line)
        if m and section:
            out[section][m.group(1)] = _parse_toml_value(m.group(2))
    return out


def _dump_toml_value(v: Any) -> str:
    if isinstance(v, bool):
        return "true" if v else "false"
    if isinstance(v, (int, float)):
        return repr(v)
    return '"' + str(v).replace("\\", "\\\\").replace('"', '\\"') + '"'


# ---------------------------------------------------------------- load/save

def load_config(path: Path | None = None, env: dict[str, str] | None = None) -> Config:
    """Load config with priority env > TOML file > defaults
    (reference: config.py:385-457)."""
    cfg = Config()
    if path is None:
        path = default_data_dir() / "config.toml"
    if path.exists():
        try:
            data = parse_toml(path.read_text(encoding="utf-8"))
        except OSError as e:
            log.warning("config: cannot read %s: %s", path, e)
            data = {}
        for sec_name, kv in data.items():
            if not hasattr(cfg, sec_name) or sec_name == "data_dir":
                log.warning("config: unknown section [%s] ignored", sec_name)
                continue
            section = getattr(cfg, sec_name)
            for key, raw in kv.items():
                section = _apply(sec_name, section, key, raw)
            cfg = dataclasses.replace(cfg, **{sec_name: section})

    env = dict(os.environ) if env is None else env
    for var, raw in env.items():
        if not var.startswith("INFOMESH_"):
            continue
        rest = var[len("INFOMESH_"):]
        if rest == "DATA_DIR":
            cfg = dataclasses.replace(
                cfg, node=dataclasses.replace(cfg.node, data_dir=raw))
            continue
        # INFOMESH_{SECTION}_{KEY}; section = first token, key = rest.
        parts = rest.lower().split("_", 1)
        if len(parts) != 2:
            continue
        sec_name, key = parts
        if not hasattr(cfg, sec_name):
            continue
        section = _apply(sec_name, getattr(cfg, sec_name), key, raw)
        cfg = dataclasses.replace(cfg, **{sec_name: section})
    return cfg


def save_config(cfg: Config, path: Path | None = None) -> Path:
    """Write only keys that differ from defaults (reference: config.py:460-525)."""
    if path is None:
        path = cfg.data_dir / "config.toml"
    default = Config()
    lines: list[str] = ["# infomesh-amd config (only non-default keys are written)"]
    for f in fields(cfg):
        sec, dsec = getattr(cfg, f.name), getattr(default, f.name)
        diff = {sf.name: getattr(sec, sf.name) for sf in fields(sec)
                if getattr(sec, sf.name) != getattr(dsec, sf.name)}
        if diff:
            lines.append(f"\n[{f.name}]")
            lines.extend(f"{k} = {_dump_toml_value(v)}" for k, v in diff.items())
    path.parent.mkdir(parents=True, exist_ok=True)
    tmp = path.with_suffix(".tmp")
    tmp.write_text("\n".join(lines) + "\n", encoding="utf-8")
    tmp.replace(path)
    return path
