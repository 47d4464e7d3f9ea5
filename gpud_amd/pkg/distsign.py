"""Release signing with ed25519 (reference: pkg/release/distsign/
distsign.go:81-183 — a root key signs signing keys, signing keys sign
release artifacts).

Pure-Python RFC 8032 ed25519 (no external crypto dependency is available
in this environment). Slow (~ms per op) but this path only runs on
self-update verification, never on the poll path.
"""

from __future__ import annotations

import hashlib
import os
from typing import Optional, Tuple

# -- ed25519 (RFC 8032) ------------------------------------------------------

_P = 2**255 - 19
_L = 2**252 + 27742317777372353535851937790883648493
_D = -121665 * pow(121666, _P - 2, _P) % _P
_I = pow(2, (_P - 1) // 4, _P)


def _sha512(m: bytes) -> bytes:
    return hashlib.sha512(m).digest()


def _inv(x: int) -> int:
    return pow(x, _P - 2, _P)


def _xrecover(y: int) -> int:
    xx = (y * y - 1) * _inv(_D * y * y + 1)
    x = pow(xx, (_P + 3) // 8, _P)
    if (x * x - xx) % _P != 0:
        x = (x * _I) % _P
    if x % 2 != 0:
        x = _P - x
    return x


_BY = 4 * _inv(5) % _P
_BX = _xrecover(_BY)
_B = (_BX % _P, _BY % _P, 1, (_BX * _BY) % _P)


def _edwards_add(p, q):
    x1, y1, z1, t1 = p
    x2, y2, z2, t2 = q
    a = (y1 - x1) * (y2 - x2) % _P
    b = (y1 + x1) * (y2 + x2) % _P
    c = t1 * 2 * _D * t2 % _P
    dd = z1 * 2 * z2 % _P
    e = b - a
    f = dd - c
    g = dd + c
    h = b + a
    return (e * f % _P, g * h % _P, f * g % _P, e * h % _P)


def _scalarmult(p, e: int):
    q = (0, 1, 1, 0)
    while e > 0:
        if e & 1:
            q = _edwards_add(q, p)
        p = _edwards_add(p, p)
        e >>= 1
    return q


def _point_compress(p) -> bytes:
    x, y, z, _t = p
    zi = _inv(z)
    x, y = x * zi % _P, y * zi % _P
    return int.to_bytes(y | ((x & 1) << 255), 32, "little")


def _point_decompress(s: bytes):
    y = int.from_bytes(s, "little")
    sign = y >> 255
    y &= (1 << 255) - 1
    if y >= _P:
        return None
    x = _xrecover(y)
    if x & 1 != sign:
        x = _P - x
    p = (x, y, 1, (x * y) % _P)
    # verify on curve
    xx, yy = x, y
    if (-xx * xx + yy * yy - 1 - _D * xx * xx * yy * yy) % _P != 0:
        return None
    return p


def generate_keypair(seed: Optional[bytes] = None) -> Tuple[bytes, bytes]:
    """Returns (private_seed32, public32)."""
    seed = seed or os.urandom(32)
    h = _sha512(seed)
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    pub = _point_compress(_scalarmult(_B, a))
    return seed, pub


def sign(message: bytes, seed: bytes) -> bytes:
    h = _sha512(seed)
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    pub = _point_compress(_scalarmult(_B, a))
    r = int.from_bytes(_sha512(h[32:] + message), "little") % _L
    rp = _point_compress(_scalarmult(_B, r))
    k = int.from_bytes(_sha512(rp + pub + message), "little") % _L
    s = (r + k * a) % _L
    return rp + int.to_bytes(s, 32, "little")


def verify(message: bytes, signature: bytes, public: bytes) -> bool:
    if len(signature) != 64 or len(public) != 32:
        return False
    rp = _point_decompress(signature[:32])
    ap = _point_decompress(public)
    if rp is None or ap is None:
        return False
    s = int.from_bytes(signature[32:], "little")
    if s >= _L:
        return False
    k = int.from_bytes(_sha512(signature[:32] + public + message), "little") % _L
    left = _scalarmult(_B, s)
    right = _edwards_add(rp, _scalarmult(ap, k))
    # compare compressed forms (projective coordinates differ)
    return _point_compress(left) == _point_compress(right)


# -- distsign chain (root key -> signing key -> artifact) --------------------

def verify_release(
    artifact: bytes,
    artifact_sig: bytes,
    signing_pub: bytes,
    signing_pub_sig: bytes,
    root_pub: bytes,
) -> bool:
    """Root key vouches for the signing key; signing key vouches for the
    artifact (reference distsign.go chain)."""
    if not verify(signing_pub, signing_pub_sig, root_pub):
        return False
    return verify(artifact, artifact_sig, signing_pub)
