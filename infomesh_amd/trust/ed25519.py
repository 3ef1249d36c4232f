"""Pure-Python Ed25519 (RFC 8032) — no crypto wheels exist in this
image, and the trust subsystem needs asymmetric signatures (reference
uses the `cryptography` package: infomesh/p2p/keys.py:41-191).

Signing rate is low (ledger entries, attestations, takedowns), so a
few ms per operation in Python is acceptable. Verified against the
RFC 8032 test vectors in tests/test_trust.py.
"""
from __future__ import annotations

import hashlib
import secrets

P = 2**255 - 19
L = 2**252 + 27742317777372353535851937790883648493
D = (-121665 * pow(121666, P - 2, P)) % P
I_SQRT = pow(2, (P - 1) // 4, P)

# Base point
_By = (4 * pow(5, P - 2, P)) % P


def _recover_x(y: int, sign: int) -> int | None:
    if y >= P:
        return None
    x2 = (y * y - 1) * pow(D * y * y + 1, P - 2, P)
    if x2 == 0:
        if sign:
            return None
        return 0
    x = pow(x2, (P + 3) // 8, P)
    if (x * x - x2) % P != 0:
        x = x * I_SQRT % P
    if (x * x - x2) % P != 0:
        return None
    if (x & 1) != sign:
        x = P - x
    return x


_Bx = _recover_x(_By, 0)
B = (_Bx, _By, 1, _Bx * _By % P)  # extended coords (X, Y, Z, T)
IDENT = (0, 1, 1, 0)


def _add(p1, p2):
    X1, Y1, Z1, T1 = p1
    X2, Y2, Z2, T2 = p2
    A = (Y1 - X1) * (Y2 - X2) % P
    Bv = (Y1 + X1) * (Y2 + X2) % P
    C = 2 * T1 * D * T2 % P
    Dv = 2 * Z1 * Z2 % P
    E, F, G, H = Bv - A, Dv - C, Dv + C, Bv + A
    return (E * F % P, G * H % P, F * G % P, E * H % P)


def _mul(s: int, pt):
    q = IDENT
    while s > 0:
        if s & 1:
            q = _add(q, pt)
        pt = _add(pt, pt)
        s >>= 1
    return q


def _compress(pt) -> bytes:
    X, Y, Z, _ = pt
    zinv = pow(Z, P - 2, P)
    x, y = X * zinv % P, Y * zinv % P
    return int.to_bytes(y | ((x & 1) << 255), 32, "little")


def _decompress(data: bytes):
    if len(data) != 32:
        return None
    y = int.from_bytes(data, "little")
    sign = y >> 255
    y &= (1 << 255) - 1
    x = _recover_x(y, sign)
    if x is None:
        return None
    return (x, y, 1, x * y % P)


def _sha512_int(*chunks: bytes) -> int:
    h = hashlib.sha512()
    for c in chunks:
        h.update(c)
    return int.from_bytes(h.digest(), "little")


def _secret_expand(seed: bytes) -> tuple[int, bytes]:
    h = hashlib.sha512(seed).digest()
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    return a, h[32:]


def generate_seed() -> bytes:
    return secrets.token_bytes(32)


def public_key(seed: bytes) -> bytes:
    a, _ = _secret_expand(seed)
    return _compress(_mul(a, B))


def sign(seed: bytes, message: bytes) -> bytes:
    a, prefix = _secret_expand(seed)
    pub = _compress(_mul(a, B))
    r = _sha512_int(prefix, message) % L
    R = _compress(_mul(r, B))
    k = _sha512_int(R, pub, message) % L
    s = (r + k * a) % L
    return R + int.to_bytes(s, 32, "little")


def verify(pub: bytes, message: bytes, signature: bytes) -> bool:
    if len(signature) != 64 or len(pub) != 32:
        return False
    A = _decompress(pub)
    if A is None:
        return False
    Rb, sb = signature[:32], signature[32:]
    R = _decompress(Rb)
    if R is None:
        return False
    s = int.from_bytes(sb, "little")
    if s >= L:
        return False
    k = _sha512_int(Rb, pub, message) % L
    left = _mul(s, B)
    right = _add(R, _mul(k, A))
    # compare affine
    lX, lY, lZ, _ = left
    rX, rY, rZ, _ = right
    return (lX * rZ - rX * lZ) % P == 0 and (lY * rZ - rY * lZ) % P == 0
