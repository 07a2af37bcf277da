"""Ed25519 signature verification (RFC 8032), pure Python.

Used for policy-bundle signature verification (oracle: safetykernel
kernel.go:787-868 — SAFETY_POLICY_PUBLIC_KEY / SAFETY_POLICY_SIGNATURE).
No third-party crypto is available offline, so this is a direct, dependency-
free implementation of the verify path (sign is included for tests).
Performance is irrelevant here: it runs once per policy (re)load.
"""
from __future__ import annotations

import hashlib

P = 2**255 - 19
L = 2**252 + 27742317777372353535851937790883648493
D = -121665 * pow(121666, P - 2, P) % P
I = pow(2, (P - 1) // 4, P)

BY = 4 * pow(5, P - 2, P) % P
BX_candidate = None


def _xrecover(y: int) -> int:
    xx = (y * y - 1) * pow(D * y * y + 1, P - 2, P)
    x = pow(xx, (P + 3) // 8, P)
    if (x * x - xx) % P != 0:
        x = x * I % P
    if x % 2 != 0:
        x = P - x
    return x


BX = _xrecover(BY)
B = (BX, BY, 1, BX * BY % P)  # extended coords
IDENT = (0, 1, 1, 0)


def _add(p, q):
    x1, y1, z1, t1 = p
    x2, y2, z2, t2 = q
    a = (y1 - x1) * (y2 - x2) % P
    b = (y1 + x1) * (y2 + x2) % P
    c = 2 * t1 * t2 * D % P
    dd = 2 * z1 * z2 % P
    e, f, g, h = b - a, dd - c, dd + c, b + a
    return (e * f % P, g * h % P, f * g % P, e * h % P)


def _scalarmult(p, e: int):
    q = IDENT
    while e > 0:
        if e & 1:
            q = _add(q, p)
        p = _add(p, p)
        e >>= 1
    return q


def _compress(p) -> bytes:
    x, y, z, _ = p
    zi = pow(z, P - 2, P)
    x, y = x * zi % P, y * zi % P
    return (y | ((x & 1) << 255)).to_bytes(32, "little")


def _decompress(s: bytes):
    y = int.from_bytes(s, "little")
    sign = y >> 255
    y &= (1 << 255) - 1
    if y >= P:
        return None
    x = _xrecover(y)
    if x & 1 != sign:
        x = P - x
    # on-curve check: -x^2 + y^2 = 1 + d x^2 y^2
    if (-x * x + y * y - 1 - D * x * x * y * y) % P != 0:
        return None
    return (x, y, 1, x * y % P)


def _h(data: bytes) -> int:
    return int.from_bytes(hashlib.sha512(data).digest(), "little")


def verify(public_key: bytes, signature: bytes, message: bytes) -> bool:
    if len(public_key) != 32 or len(signature) != 64:
        return False
    a = _decompress(public_key)
    if a is None:
        return False
    rs = signature[:32]
    r = _decompress(rs)
    if r is None:
        return False
    s = int.from_bytes(signature[32:], "little")
    if s >= L:
        return False
    k = _h(rs + public_key + message) % L
    left = _scalarmult(B, s)
    right = _add(r, _scalarmult(a, k))
    return _compress(left) == _compress(right)


# -- signing (tests only) ----------------------------------------------------


def _secret_expand(secret: bytes):
    h = hashlib.sha512(secret).digest()
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    return a, h[32:]


def public_key(secret: bytes) -> bytes:
    a, _ = _secret_expand(secret)
    return _compress(_scalarmult(B, a))


def sign(secret: bytes, message: bytes) -> bytes:
    a, prefix = _secret_expand(secret)
    pk = _compress(_scalarmult(B, a))
    r = _h(prefix + message) % L
    rp = _compress(_scalarmult(B, r))
    k = _h(rp + pk + message) % L
    s = (r + k * a) % L
    return rp + s.to_bytes(32, "little")
