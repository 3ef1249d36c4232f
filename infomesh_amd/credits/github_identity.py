"""Owner identity from git configuration.

Reference parity: infomesh/credits/github_identity.py (git-email
detection + first-start checks). The email never leaves the machine:
only its hash (credits/sync.owner_hash) is shared.
"""
from __future__ import annotations

import re
import subprocess
from pathlib import Path

_EMAIL_RE = re.compile(r"^[^@\s]+@[^@\s]+\.[^@\s]+$")


def detect_git_email() -> str | None:
    try:
        out = subprocess.run(["git", "config", "--get", "user.email"],
                             capture_output=True, text=True, timeout=5)
    except (OSError, subprocess.TimeoutExpired):
        return None
    email = out.stdout.strip()
    return email if out.returncode == 0 and _EMAIL_RE.match(email) else None


def stored_owner_email(data_dir: Path) -> str | None:
    f = data_dir / "owner_email"
    if f.exists():
        email = f.read_text().strip()
        return email if _EMAIL_RE.match(email) else None
    return None


def ensure_owner_identity(data_dir: Path,
                          email: str | None = None) -> str | None:
    """First-start identity: explicit email > stored > git config.
    Stored locally (0600); only the hash is ever exported."""
    chosen = email or stored_owner_email(data_dir) or detect_git_email()
    if chosen is None or not _EMAIL_RE.match(chosen):
        return None
    data_dir.mkdir(parents=True, exist_ok=True)
    f = data_dir / "owner_email"
    if not f.exists() or f.read_text().strip() != chosen:
        f.write_text(chosen)
        f.chmod(0o600)
    return chosen
