"""Ed25519 (RFC 8032) — pure-Python sign/verify.

Discord signs interaction webhooks with Ed25519
(X-Signature-Ed25519/X-Signature-Timestamp); the reference's Discord
trigger verifies them (api/pkg/trigger/discord). No crypto package is
installed offline, and Ed25519 verification is ~60 lines of field
arithmetic, so it lives here. Performance is irrelevant at webhook
rates (~1 ms/verify).
"""
from __future__ import annotations

import hashlib

__all__ = ["verify", "sign", "public_from_secret"]

_P = 2**255 - 19
_L = 2**252 + 27742317777372353535851937790883648493
_D = (-121665 * pow(121666, _P - 2, _P)) % _P
_I = pow(2, (_P - 1) // 4, _P)


def _xrecover(y: int) -> int:
    xx = (y * y - 1) * pow(_D * y * y + 1, _P - 2, _P)
    x = pow(xx, (_P + 3) // 8, _P)
    if (x * x - xx) % _P != 0:
        x = (x * _I) % _P
    if x % 2 != 0:
        x = _P - x
    return x


_BY = 4 * pow(5, _P - 2, _P) % _P
_BX = _xrecover(_BY)
_B = (_BX, _BY, 1, (_BX * _BY) % _P)       # extended coords
_IDENT = (0, 1, 1, 0)


def _add(p, q):
    x1, y1, z1, t1 = p
    x2, y2, z2, t2 = q
    a = (y1 - x1) * (y2 - x2) % _P
    b = (y1 + x1) * (y2 + x2) % _P
    c = 2 * t1 * t2 * _D % _P
    d = 2 * z1 * z2 % _P
    e, f, g, h = b - a, d - c, d + c, b + a
    return (e * f % _P, g * h % _P, f * g % _P, e * h % _P)


def _mul(p, n: int):
    q = _IDENT
    while n:
        if n & 1:
            q = _add(q, p)
        p = _add(p, p)
        n >>= 1
    return q


def _compress(p) -> bytes:
    x, y, z, _ = p
    zi = pow(z, _P - 2, _P)
    x, y = x * zi % _P, y * zi % _P
    return int.to_bytes(y | ((x & 1) << 255), 32, "little")


def _decompress(s: bytes):
    y = int.from_bytes(s, "little")
    sign = y >> 255
    y &= (1 << 255) - 1
    if y >= _P:
        raise ValueError("bad point")
    x = _xrecover(y)
    if x & 1 != sign:
        x = _P - x
    p = (x, y, 1, (x * y) % _P)
    # validate the point is on the curve
    if (-x * x + y * y - 1 - _D * x * x * y * y) % _P != 0:
        raise ValueError("point not on curve")
    return p


def _sha512_int(*parts: bytes) -> int:
    return int.from_bytes(hashlib.sha512(b"".join(parts)).digest(),
                          "little")


def public_from_secret(secret: bytes) -> bytes:
    h = hashlib.sha512(secret).digest()
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    return _compress(_mul(_B, a))


def sign(message: bytes, secret: bytes) -> bytes:
    h = hashlib.sha512(secret).digest()
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    pub = _compress(_mul(_B, a))
    r = _sha512_int(h[32:], message) % _L
    rp = _compress(_mul(_B, r))
    k = _sha512_int(rp, pub, message) % _L
    s = (r + k * a) % _L
    return rp + int.to_bytes(s, 32, "little")


def verify(message: bytes, signature: bytes, public_key: bytes) -> bool:
    if len(signature) != 64 or len(public_key) != 32:
        return False
    try:
        a = _decompress(public_key)
        rp = signature[:32]
        _decompress(rp)                 # R must be a valid point
    except ValueError:
        return False
    s = int.from_bytes(signature[32:], "little")
    if s >= _L:
        return False
    k = _sha512_int(rp, public_key, message) % _L
    left = _mul(_B, s)
    right = _add(_decompress(rp), _mul(a, k))
    return _compress(left) == _compress(right)
