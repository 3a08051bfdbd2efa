"""Minimal HS256 JWT (stdlib only — no PyJWT in the image).

Implements exactly what the reference API uses (reference api.py:318-361):
``create_access_token`` signing ``{"sub": username, "exp": ...}`` and a
decode that validates signature + expiry.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Any, Dict


class JWTError(Exception):
    pass


def _b64url(data: bytes) -> bytes:
    return base64.urlsafe_b64encode(data).rstrip(b"=")


def _b64url_decode(data: str) -> bytes:
    pad = -len(data) % 4
    return base64.urlsafe_b64decode(data + "=" * pad)


def encode(payload: Dict[str, Any], secret: str, algorithm: str = "HS256") -> str:
    if algorithm != "HS256":
        raise JWTError(f"unsupported algorithm {algorithm}")
    header = _b64url(json.dumps({"alg": "HS256", "typ": "JWT"},
                                separators=(",", ":")).encode())
    body = _b64url(json.dumps(payload, separators=(",", ":")).encode())
    signing_input = header + b"." + body
    sig = hmac.new(secret.encode(), signing_input, hashlib.sha256).digest()
    return (signing_input + b"." + _b64url(sig)).decode()


def decode(token: str, secret: str, algorithm: str = "HS256") -> Dict[str, Any]:
    if algorithm != "HS256":
        raise JWTError(f"unsupported algorithm {algorithm}")
    try:
        header_b64, body_b64, sig_b64 = token.split(".")
    except ValueError:
        raise JWTError("malformed token") from None
    signing_input = (header_b64 + "." + body_b64).encode()
    expected = hmac.new(secret.encode(), signing_input, hashlib.sha256).digest()
    if not hmac.compare_digest(expected, _b64url_decode(sig_b64)):
        raise JWTError("signature mismatch")
    try:
        payload = json.loads(_b64url_decode(body_b64))
    except Exception:
        raise JWTError("malformed payload") from None
    exp = payload.get("exp")
    if exp is not None and time.time() > float(exp):
        raise JWTError("token expired")
    return payload
