"""Port availability + environment detection.

Reference parity: infomesh/resources/port_check.py (bind checks, WSL
detection, cloud-provider hints, fix suggestions) — trimmed to what a
single-node deployment needs; the reference's 1200-line cloud-NSG
autofix machinery is out of scope for an offline node.
"""
from __future__ import annotations

import socket
from pathlib import Path


def port_available(port: int, host: str = "127.0.0.1") -> bool:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        try:
            s.bind((host, port))
            return True
        except OSError:
            return False


def find_free_port(start: int = 8080, tries: int = 50) -> int | None:
    for p in range(start, start + tries):
        if port_available(p):
            return p
    return None


def is_wsl() -> bool:
    try:
        return "microsoft" in Path("/proc/version").read_text().lower()
    except OSError:
        return False


def detect_environment() -> dict:
    env = {"wsl": is_wsl(), "container": Path("/.dockerenv").exists()}
    try:
        vendor = Path("/sys/class/dmi/id/sys_vendor").read_text().strip()
    except OSError:
        vendor = ""
    env["cloud"] = ("azure" if "Microsoft" in vendor else
                    "gcp" if "Google" in vendor else
                    "aws" if "Amazon" in vendor else
                    vendor or "unknown")
    return env


def check_port_with_advice(port: int) -> dict:
    ok = port_available(port)
    out = {"port": port, "available": ok, "env": detect_environment()}
    if not ok:
        alt = find_free_port(port + 1)
        out["advice"] = (f"port {port} is in use — another node running? "
                         f"try --port {alt}" if alt else
                         f"port {port} and the next 50 are all in use")
    return out
