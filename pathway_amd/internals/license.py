"""License keys & entitlements (reference src/engine/license.rs:28-60 +
src/engine/dataflow/config.rs:11-15).

The reference verifies ed25519-signed license keys and gates features
("unlimited-workers", MONITORING, ...) plus the MAX_WORKERS=8 cap.  This
build implements the same scheme with a pure-python RFC 8032 ed25519
verifier (no crypto dependency): a key is

    base64( payload-json || 64-byte signature )

signed over the payload by the issuer key.  Without a valid key the
engine runs free-tier: all core features, worker count capped at 8.
`issue_key` (signing) exists so deployments and tests can mint keys for
their own issuer pair.
"""

from __future__ import annotations

import base64
import hashlib
import json
from typing import Any

# ---------------------------------------------------------------------------
# RFC 8032 ed25519 (pure python, compact)
# ---------------------------------------------------------------------------

_P = 2**255 - 19
_L = 2**252 + 27742317777372353535851937790883648493
_D = (-121665 * pow(121666, _P - 2, _P)) % _P
_I = pow(2, (_P - 1) // 4, _P)


def _sha512(b: bytes) -> bytes:
    return hashlib.sha512(b).digest()


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
_B = (_BX, _BY, 1, (_BX * _BY) % _P)  # extended coords


def _edwards_add(p, q):
    x1, y1, z1, t1 = p
    x2, y2, z2, t2 = q
    a = (y1 - x1) * (y2 - x2) % _P
    b = (y1 + x1) * (y2 + x2) % _P
    c = 2 * t1 * t2 * _D % _P
    dd = 2 * z1 * z2 % _P
    e, f, g, h = b - a, dd - c, dd + c, b + a
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
    x, y, z, _ = p
    zi = _inv(z)
    x, y = x * zi % _P, y * zi % _P
    return int.to_bytes(y | ((x & 1) << 255), 32, "little")


def _point_decompress(s: bytes):
    y = int.from_bytes(s, "little")
    sign = y >> 255
    y &= (1 << 255) - 1
    x = _xrecover(y)
    if x & 1 != sign:
        x = _P - x
    if (-x * x + y * y - 1 - _D * x * x * y * y) % _P != 0:
        raise ValueError("invalid point")
    return (x, y, 1, (x * y) % _P)


def _point_equal(p, q) -> bool:
    x1, y1, z1, _ = p
    x2, y2, z2, _ = q
    return (x1 * z2 - x2 * z1) % _P == 0 and (y1 * z2 - y2 * z1) % _P == 0


def ed25519_public_key(seed: bytes) -> bytes:
    h = _sha512(seed)
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    return _point_compress(_scalarmult(_B, a))


def ed25519_sign(seed: bytes, msg: bytes) -> bytes:
    h = _sha512(seed)
    a = int.from_bytes(h[:32], "little")
    a &= (1 << 254) - 8
    a |= 1 << 254
    prefix = h[32:]
    pub = _point_compress(_scalarmult(_B, a))
    r = int.from_bytes(_sha512(prefix + msg), "little") % _L
    rp = _point_compress(_scalarmult(_B, r))
    k = int.from_bytes(_sha512(rp + pub + msg), "little") % _L
    s = (r + k * a) % _L
    return rp + int.to_bytes(s, 32, "little")


def ed25519_verify(pub: bytes, msg: bytes, sig: bytes) -> bool:
    if len(sig) != 64 or len(pub) != 32:
        return False
    try:
        a = _point_decompress(pub)
        rp = _point_decompress(sig[:32])
    except ValueError:
        return False
    s = int.from_bytes(sig[32:], "little")
    if s >= _L:
        return False
    k = int.from_bytes(_sha512(sig[:32] + pub + msg), "little") % _L
    return _point_equal(
        _scalarmult(_B, s), _edwards_add(rp, _scalarmult(a, k))
    )


# ---------------------------------------------------------------------------
# licensing
# ---------------------------------------------------------------------------

#: default issuer (a fixed development keypair; deployments set their own
#: via PATHWAY_LICENSE_ISSUER_PUBKEY)
_DEV_SEED = hashlib.sha256(b"pathway_amd development issuer").digest()
DEV_ISSUER_PUBLIC_KEY = ed25519_public_key(_DEV_SEED)

#: reference config.rs:11-15 — worker cap without unlimited-workers
MAX_WORKERS_FREE = 8

ENTITLEMENT_UNLIMITED_WORKERS = "unlimited-workers"
ENTITLEMENT_MONITORING = "monitoring"


class License:
    def __init__(self, payload: dict[str, Any] | None, valid: bool):
        self.payload = payload or {}
        self.valid = valid

    @property
    def entitlements(self) -> set[str]:
        if not self.valid:
            return set()
        return set(self.payload.get("entitlements", []))

    def has(self, entitlement: str) -> bool:
        return entitlement in self.entitlements

    def max_workers(self) -> int | None:
        """None = unlimited."""
        if self.has(ENTITLEMENT_UNLIMITED_WORKERS):
            return None
        return MAX_WORKERS_FREE

    def __repr__(self) -> str:
        kind = self.payload.get("tier", "free") if self.valid else "free"
        return f"<License {kind} entitlements={sorted(self.entitlements)}>"


FREE = License(None, False)


def issue_key(entitlements: list[str], *, tier: str = "enterprise",
              issuer_seed: bytes = _DEV_SEED, **extra: Any) -> str:
    payload = json.dumps(
        {"tier": tier, "entitlements": entitlements, **extra},
        sort_keys=True,
    ).encode()
    sig = ed25519_sign(issuer_seed, payload)
    return base64.b64encode(payload + sig).decode()


def parse_key(key: str | None,
              issuer_public_key: bytes = DEV_ISSUER_PUBLIC_KEY) -> License:
    if not key:
        return FREE
    try:
        raw = base64.b64decode(key.strip())
        payload, sig = raw[:-64], raw[-64:]
        if not ed25519_verify(issuer_public_key, payload, sig):
            return FREE
        return License(json.loads(payload), True)
    except Exception:
        return FREE


def check_worker_limit(workers: int, key: str | None) -> None:
    """Raise if the worker count exceeds the license cap
    (reference dataflow/config.rs:11-15)."""
    lic = parse_key(key)
    cap = lic.max_workers()
    if cap is not None and workers > cap:
        raise RuntimeError(
            f"{workers} workers requested but the license allows at most "
            f"{cap}; an '{ENTITLEMENT_UNLIMITED_WORKERS}' entitlement is "
            "required for more (reference MAX_WORKERS cap)"
        )
