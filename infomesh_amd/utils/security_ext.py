"""Extended security controls for the admin surfaces.

Reference parity: infomesh/security_ext.py (TLS/JWT/RBAC/IP-filter/
webhook HMAC). Dependency-free: tokens are HS256-style HMAC-signed
(stdlib hmac/hashlib), RBAC is a static role->operations map, webhook
payloads carry an `X-Infomesh-Signature` HMAC header. TLS termination
is delegated to the reverse proxy in front of the localhost-only API
(the reference's in-process TLS served the same localhost surface).
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import ipaddress
import json
import time

# ----------------------------------------------------------------- RBAC

ROLES: dict[str, set[str]] = {
    "admin": {"search", "crawl", "index", "config", "keys", "metrics",
              "credits", "compliance"},
    "operator": {"search", "crawl", "index", "metrics", "credits"},
    "reader": {"search", "metrics"},
}


def role_allows(role: str, operation: str) -> bool:
    return operation in ROLES.get(role, set())


# ------------------------------------------------------------ IP filter

class IpFilter:
    """Allow/deny lists of CIDR networks; deny wins; default allow."""

    def __init__(self, allow: list[str] | None = None,
                 deny: list[str] | None = None):
        self.allow = [ipaddress.ip_network(a) for a in (allow or [])]
        self.deny = [ipaddress.ip_network(d) for d in (deny or [])]

    def permitted(self, addr: str) -> bool:
        try:
            ip = ipaddress.ip_address(addr)
        except ValueError:
            return False
        if any(ip in n for n in self.deny):
            return False
        if self.allow:
            return any(ip in n for n in self.allow)
        return True


# --------------------------------------------------------- signed token

def _b64(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _unb64(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def issue_token(secret: bytes, subject: str, role: str,
                ttl_s: float = 3600.0, now=time.time) -> str:
    """HS256-style compact token: b64(payload).b64(hmac)."""
    payload = json.dumps({"sub": subject, "role": role,
                          "exp": now() + ttl_s},
                         separators=(",", ":")).encode()
    sig = hmac.new(secret, payload, hashlib.sha256).digest()
    return f"{_b64(payload)}.{_b64(sig)}"


def verify_token(secret: bytes, token: str,
                 now=time.time) -> dict | None:
    """Returns the payload dict, or None if forged/expired."""
    try:
        p64, s64 = token.split(".", 1)
        payload = _unb64(p64)
        sig = _unb64(s64)
    except (ValueError, TypeError):
        return None
    good = hmac.new(secret, payload, hashlib.sha256).digest()
    if not hmac.compare_digest(sig, good):
        return None
    try:
        data = json.loads(payload)
    except ValueError:
        return None
    if data.get("exp", 0) < now():
        return None
    return data


# -------------------------------------------------------- webhook HMAC

SIGNATURE_HEADER = "X-Infomesh-Signature"


def sign_webhook(secret: bytes, body: bytes) -> str:
    """Value for X-Infomesh-Signature: sha256=<hex hmac>."""
    return "sha256=" + hmac.new(secret, body, hashlib.sha256).hexdigest()


def verify_webhook(secret: bytes, body: bytes, header: str) -> bool:
    return hmac.compare_digest(sign_webhook(secret, body), header or "")
