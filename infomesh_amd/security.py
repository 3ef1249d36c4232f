"""SSRF guard for crawler fetches.

Reference parity: infomesh/security.py:67-168 (validate_url blocks
private/link-local IPs and bad schemes, optional DNS resolution;
validate_url_post_redirect re-checks after redirects).
"""
from __future__ import annotations

import ipaddress
import socket
from urllib.parse import urlparse

from .errors import InfoMeshError

ALLOWED_SCHEMES = ("http", "https")
BLOCKED_HOSTS = ("localhost", "metadata.google.internal", "169.254.169.254")
MAX_URL_LEN = 4096


def _ip_is_blocked(ip: ipaddress.IPv4Address | ipaddress.IPv6Address) -> bool:
    return (ip.is_private or ip.is_loopback or ip.is_link_local
            or ip.is_multicast or ip.is_reserved or ip.is_unspecified)


def validate_url(url: str, resolve_dns: bool = False) -> str:
    """Validate a URL for crawling; returns the URL or raises CRWL001.

    Blocks: non-http(s) schemes, missing host, localhost aliases,
    literal private/link-local/multicast IPs, userinfo tricks, and
    (optionally, with DNS) hosts resolving to private addresses.
    """
    if not url or len(url) > MAX_URL_LEN:
        raise InfoMeshError("CRWL001", "empty or oversized URL")
    try:
        parsed = urlparse(url)
    except ValueError as e:
        raise InfoMeshError("CRWL001", f"unparseable URL: {e}") from e
    if parsed.scheme not in ALLOWED_SCHEMES:
        raise InfoMeshError("CRWL001", f"scheme {parsed.scheme!r} not allowed")
    host = parsed.hostname
    if not host:
        raise InfoMeshError("CRWL001", "no host")
    if "@" in parsed.netloc:
        raise InfoMeshError("CRWL001", "userinfo in URL")
    host_l = host.lower().rstrip(".")
    if host_l in BLOCKED_HOSTS or host_l.endswith(".localhost"):
        raise InfoMeshError("CRWL001", f"blocked host {host!r}")
    try:
        ip = ipaddress.ip_address(host_l)
        if _ip_is_blocked(ip):
            raise InfoMeshError("CRWL001", f"blocked IP {host!r}")
    except ValueError:
        # Not an IP literal — optionally resolve.
        if resolve_dns:
            try:
                infos = socket.getaddrinfo(host_l, None)
            except OSError as e:
                raise InfoMeshError("CRWL001", f"DNS resolution failed: {e}") from e
            for info in infos:
                addr = info[4][0]
                try:
                    if _ip_is_blocked(ipaddress.ip_address(addr)):
                        raise InfoMeshError(
                            "CRWL001", f"{host!r} resolves to blocked {addr}")
                except ValueError:
                    continue
    return url


def validate_url_post_redirect(url: str, resolve_dns: bool = True) -> str:
    """Re-validate after an HTTP redirect (reference: security.py:129)."""
    return validate_url(url, resolve_dns=resolve_dns)


def is_url_safe(url: str, resolve_dns: bool = False) -> bool:
    try:
        validate_url(url, resolve_dns=resolve_dns)
        return True
    except InfoMeshError:
        return False
