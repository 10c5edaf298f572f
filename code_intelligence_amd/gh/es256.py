"""Self-contained ES256 (ECDSA P-256 / SHA-256) JWT signing.

Reference parity: chatbot/cmd/jwt/main.go signs an ES256 JWT for testing
the ISTIO-protected chatbot webhook. No EC library is in the MI355X
image, so this implements the minimum: SEC1/PKCS#8 P-256 private-key PEM
parsing and RFC 6979-style deterministic ECDSA signatures in the raw
r||s JWT form. Verified against `openssl dgst -sha256 -verify`.
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import re
import time
from typing import Tuple

from .rs256 import _read_tlv  # minimal DER reader

# NIST P-256 domain parameters
P = 0xFFFFFFFF00000001000000000000000000000000FFFFFFFFFFFFFFFFFFFFFFFF
A = P - 3
B = 0x5AC635D8AA3A93E7B3EBBD55769886BC651D06B0CC53B0F63BCE3C3E27D2604B
N = 0xFFFFFFFF00000000FFFFFFFFFFFFFFFFBCE6FAADA7179E84F3B9CAC2FC632551
GX = 0x6B17D1F2E12C4247F8BCE6E563A440F277037D812DEB33A0F4A13945D898C296
GY = 0x4FE342E2FE1A7F9B8EE7EB4A7C0F9E162BCE33576B315ECECBB6406837BF51F5


def _inv(a: int, m: int) -> int:
    return pow(a, m - 2, m)


def _point_add(p1, p2):
    if p1 is None:
        return p2
    if p2 is None:
        return p1
    x1, y1 = p1
    x2, y2 = p2
    if x1 == x2 and (y1 + y2) % P == 0:
        return None
    if p1 == p2:
        lam = (3 * x1 * x1 + A) * _inv(2 * y1, P) % P
    else:
        lam = (y2 - y1) * _inv((x2 - x1) % P, P) % P
    x3 = (lam * lam - x1 - x2) % P
    y3 = (lam * (x1 - x3) - y1) % P
    return (x3, y3)


def _point_mul(k: int, point):
    result = None
    addend = point
    while k:
        if k & 1:
            result = _point_add(result, addend)
        addend = _point_add(addend, addend)
        k >>= 1
    return result


def parse_ec_private_key_pem(pem: str) -> int:
    """Private scalar d from a SEC1 'EC PRIVATE KEY' or PKCS#8 PEM."""
    m = re.search(
        r"-----BEGIN (EC )?PRIVATE KEY-----(.*?)-----END (EC )?PRIVATE KEY-----",
        pem, re.S)
    if not m:
        raise ValueError("no EC private key PEM found")
    der = base64.b64decode(re.sub(r"\s", "", m.group(2)))
    tag, seq, _ = _read_tlv(der, 0)
    assert tag == 0x30
    # SEC1: SEQ{ INTEGER 1, OCTET STRING d, [0] params, [1] pubkey }
    # PKCS#8: SEQ{ INTEGER 0, SEQ alg, OCTET STRING{ SEC1 } }
    t0, v0, i = _read_tlv(seq, 0)
    version = int.from_bytes(v0, "big")
    t1, v1, i = _read_tlv(seq, i)
    if version == 0 and t1 == 0x30:      # PKCS#8 wrapper
        t2, inner, _ = _read_tlv(seq, i)  # OCTET STRING
        assert t2 == 0x04
        tag, seq, _ = _read_tlv(inner, 0)
        t0, v0, i = _read_tlv(seq, 0)
        t1, v1, i = _read_tlv(seq, i)
    assert t1 == 0x04, "expected OCTET STRING private scalar"
    return int.from_bytes(v1, "big")


def _rfc6979_k(d: int, h: bytes) -> int:
    """Deterministic nonce (RFC 6979, SHA-256, qlen=256)."""
    x = d.to_bytes(32, "big")
    v = b"\x01" * 32
    k = b"\x00" * 32
    k = hmac.new(k, v + b"\x00" + x + h, hashlib.sha256).digest()
    v = hmac.new(k, v, hashlib.sha256).digest()
    k = hmac.new(k, v + b"\x01" + x + h, hashlib.sha256).digest()
    v = hmac.new(k, v, hashlib.sha256).digest()
    while True:
        v = hmac.new(k, v, hashlib.sha256).digest()
        cand = int.from_bytes(v, "big")
        if 1 <= cand < N:
            return cand
        k = hmac.new(k, v + b"\x00", hashlib.sha256).digest()
        v = hmac.new(k, v, hashlib.sha256).digest()


def sign_es256(message: bytes, d: int) -> Tuple[int, int]:
    h = hashlib.sha256(message).digest()
    z = int.from_bytes(h, "big")
    k = _rfc6979_k(d, h)
    x, _ = _point_mul(k, (GX, GY))
    r = x % N
    s = _inv(k, N) * (z + r * d) % N
    if s > N // 2:  # low-s normalization (accepted by verifiers)
        s = N - s
    return r, s


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def encode_es256_jwt(payload: dict, pem: str) -> str:
    """JWT with raw r||s signature (JWS ES256 form, as the Go helper emits)."""
    d = parse_ec_private_key_pem(pem)
    header = _b64url(json.dumps({"alg": "ES256", "typ": "JWT"}).encode())
    body = _b64url(json.dumps(payload).encode())
    signing_input = f"{header}.{body}".encode()
    r, s = sign_es256(signing_input, d)
    sig = r.to_bytes(32, "big") + s.to_bytes(32, "big")
    return f"{header}.{body}.{_b64url(sig)}"


def chatbot_test_jwt(pem: str, audience: str, issuer: str = "kubeflow-chatbot",
                     ttl_s: int = 3600) -> str:
    """The chatbot/cmd/jwt/main.go payload shape."""
    now = int(time.time())
    return encode_es256_jwt({"iss": issuer, "aud": audience,
                             "iat": now, "exp": now + ttl_s}, pem)
