"""Error catalog (reference parity: infomesh/errors.py:12-197).

Structured error categories + a registry so every user-facing failure has a
stable code, a message and a hint.
"""
from __future__ import annotations

import enum
from dataclasses import dataclass


class ErrorCategory(enum.Enum):
    CONFIG = "config"
    INDEX = "index"
    SEARCH = "search"
    CRAWL = "crawl"
    NETWORK = "network"
    GPU = "gpu"
    CREDITS = "credits"
    TRUST = "trust"
    RUNTIME = "runtime"
    SECURITY = "security"


@dataclass(frozen=True)
class ErrorInfo:
    code: str
    category: ErrorCategory
    message: str
    hint: str = ""


class InfoMeshError(Exception):
    """Base exception carrying a stable error code."""

    def __init__(self, code: str, detail: str = ""):
        info = ERROR_REGISTRY.get(code)
        self.code = code
        self.info = info
        self.detail = detail
        msg = info.message if info else code
        if detail:
            msg = f"{msg}: {detail}"
        super().__init__(msg)


class GpuExtensionMissing(InfoMeshError):
    """Raised when a GPU is present but the HIP extension is not importable.

    The GPU path must never silently fall back to eager PyTorch."""

    def __init__(self, detail: str = ""):
        super().__init__("GPU001", detail)


ERROR_REGISTRY: dict[str, ErrorInfo] = {}


def _register(code: str, category: ErrorCategory, message: str, hint: str = "") -> None:
    ERROR_REGISTRY[code] = ErrorInfo(code, category, message, hint)


_register("CFG001", ErrorCategory.CONFIG, "invalid configuration value",
          "check ~/.infomesh/config.toml or INFOMESH_* env vars")
_register("CFG002", ErrorCategory.CONFIG, "unknown configuration key")
_register("IDX001", ErrorCategory.INDEX, "document store unavailable")
_register("IDX002", ErrorCategory.INDEX, "snapshot file is corrupt or exceeds import limits",
          "snapshot imports are capped at 1 GB / 100k docs")
_register("IDX003", ErrorCategory.INDEX, "unsupported FTS tokenizer",
          "allowed: unicode61, ascii, porter, trigram")
_register("SRCH001", ErrorCategory.SEARCH, "query could not be parsed")
_register("SRCH002", ErrorCategory.SEARCH, "search backend unavailable")
_register("CRWL001", ErrorCategory.CRAWL, "URL failed SSRF validation",
          "private/link-local addresses and non-http(s) schemes are blocked")
_register("CRWL002", ErrorCategory.CRAWL, "robots.txt disallows this URL")
_register("CRWL003", ErrorCategory.CRAWL, "fetch failed after retries")
_register("NET001", ErrorCategory.NETWORK, "distributed fabric not initialized",
          "searching in local-only degraded mode")
_register("GPU001", ErrorCategory.GPU, "HIP extension not built but a GPU is present",
          "run python setup.py build_ext --inplace (gfx950)")
_register("GPU002", ErrorCategory.GPU, "GPU shard out of capacity")
_register("CRD001", ErrorCategory.CREDITS, "insufficient credits for search")
_register("TRST001", ErrorCategory.TRUST, "attestation signature invalid")
_register("RT001", ErrorCategory.RUNTIME, "another instance is already running",
          "check the PID file under the data directory")
_register("SEC001", ErrorCategory.SECURITY, "request rejected by security policy")


def format_error(exc: Exception) -> str:
    """Render an exception with its code/category/hint when registered."""
    if isinstance(exc, InfoMeshError) and exc.info is not None:
        out = f"[{exc.info.code}/{exc.info.category.value}] {exc.info.message}"
        if exc.detail:
            out += f": {exc.detail}"
        if exc.info.hint:
            out += f" (hint: {exc.info.hint})"
        return out
    return f"[unregistered] {type(exc).__name__}: {exc}"
