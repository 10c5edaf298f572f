"""Self-contained RS256 (RSASSA-PKCS1-v1_5 / SHA-256) JWT signing.

The reference signs GitHub App JWTs via pyjwt+cryptography
(py/code_intelligence/github_app.py:106-119). Neither library is in the
MI355X runtime image, and GitHub App auth is control-path, so this module
implements the minimum: PEM (PKCS#1 'RSA PRIVATE KEY' or PKCS#8
'PRIVATE KEY') DER parsing and PKCS1-v1.5 SHA-256 signatures with plain
modular exponentiation. Verified against `openssl dgst -sha256 -sign`.

Caveat (documented, accepted): Python ``pow`` is not constant-time, so
signing leaks timing information about the private exponent. This path
only ever runs against an operator-supplied App key on the control path;
when the ``cryptography`` package is available, prefer it.
"""
from __future__ import annotations

import base64
import hashlib
import json
import re
import time
from typing import Dict, Tuple

# DigestInfo prefix for SHA-256 (RFC 8017 §9.2)
_SHA256_PREFIX = bytes.fromhex("3031300d060960864801650304020105000420")


def _read_tlv(data: bytes, i: int) -> Tuple[int, bytes, int]:
    tag = data[i]
    i += 1
    ln = data[i]
    i += 1
    if ln & 0x80:
        nb = ln & 0x7F
        ln = int.from_bytes(data[i: i + nb], "big")
        i += nb
    return tag, data[i: i + ln], i + ln


def _read_int(data: bytes, i: int) -> Tuple[int, int]:
    tag, content, nxt = _read_tlv(data, i)
    if tag != 0x02:
        raise ValueError(f"malformed key: expected INTEGER, got tag {tag:#x}")
    return int.from_bytes(content, "big"), nxt


def parse_rsa_private_key_pem(pem: str) -> Dict[str, int]:
    """Returns {'n', 'e', 'd'} from a PKCS#1 or PKCS#8 RSA private key PEM."""
    m = re.search(
        r"-----BEGIN (RSA )?PRIVATE KEY-----(.*?)-----END (RSA )?PRIVATE KEY-----",
        pem, re.S)
    if not m:
        raise ValueError("no PEM private key found")
    der = base64.b64decode(re.sub(r"\s", "", m.group(2)))
    tag, seq, _ = _read_tlv(der, 0)
    if tag != 0x30:
        raise ValueError("malformed key: outer DER tag is not SEQUENCE")
    # PKCS#8 wraps a PKCS#1 blob inside an OCTET STRING after an
    # AlgorithmIdentifier; PKCS#1 starts with INTEGER version then n.
    i = 0
    version, i = _read_int(seq, i)
    nxt_tag = seq[i]
    if nxt_tag == 0x30:  # PKCS#8: AlgorithmIdentifier SEQUENCE
        _, _, i = _read_tlv(seq, i)          # skip algorithm id
        tag, octet, _ = _read_tlv(seq, i)    # OCTET STRING with PKCS#1
        if tag != 0x04:
            raise ValueError("malformed PKCS#8 key: expected OCTET STRING")
        tag, seq, _ = _read_tlv(octet, 0)
        if tag != 0x30:
            raise ValueError("malformed PKCS#8 key: inner tag is not SEQUENCE")
        i = 0
        version, i = _read_int(seq, i)
    n, i = _read_int(seq, i)
    e, i = _read_int(seq, i)
    d, i = _read_int(seq, i)
    return {"n": n, "e": e, "d": d}


def sign_pkcs1_sha256(message: bytes, key: Dict[str, int]) -> bytes:
    n, d = key["n"], key["d"]
    k = (n.bit_length() + 7) // 8
    digest = hashlib.sha256(message).digest()
    t = _SHA256_PREFIX + digest
    ps = b"\xff" * (k - len(t) - 3)
    em = b"\x00\x01" + ps + b"\x00" + t
    sig = pow(int.from_bytes(em, "big"), d, n)
    return sig.to_bytes(k, "big")


def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def encode_rs256_jwt(payload: dict, pem: str) -> str:
    key = parse_rsa_private_key_pem(pem)
    header = _b64url(json.dumps({"alg": "RS256", "typ": "JWT"}).encode())
    body = _b64url(json.dumps(payload).encode())
    signing_input = f"{header}.{body}".encode()
    sig = sign_pkcs1_sha256(signing_input, key)
    return f"{header}.{body}.{_b64url(sig)}"


def app_jwt(app_id: str, pem: str, ttl_s: int = 60, now: float | None = None) -> str:
    """GitHub App JWT (reference: 60 s expiry, github_app.py:106-119)."""
    t = int(now if now is not None else time.time())
    return encode_rs256_jwt({"iat": t - 10, "exp": t + ttl_s, "iss": app_id}, pem)
