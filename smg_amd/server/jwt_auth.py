"""JWT validation + JWKS provider (reference: crates/auth/src/{jwt,jwks}.rs —
JwtValidator with algorithm-confusion guard, iss/aud/exp/nbf validation with
leeway, optional JTI replay cache, role extraction via role_claim + mapping;
JwksProvider with TTL cache and OIDC discovery).

The image has no crypto library, so signature verification is pure Python:
RS256/384/512 is PKCS#1 v1.5 (one modular exponentiation + padding check),
ES256/384 is ECDSA over NIST P-256/P-384, HS256/384/512 is stdlib hmac.
Verification-side RSA/ECDSA needs only public-key math — `pow(s, e, n)` and
curve point arithmetic — so no key-generation or secret handling lives here
beyond HMAC.  JWKS sources: inline dict, local file path, or http(s) URI
(fetched with aiohttp when egress exists; TTL-cached like jwks.rs:355-418).
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple


class JwtError(Exception):
    pass


def _b64url_decode(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def _b64url_encode(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).decode().rstrip("=")


def _b64url_to_int(s: str) -> int:
    return int.from_bytes(_b64url_decode(s), "big")


# ---- RSA PKCS#1 v1.5 verify -------------------------------------------------
# DigestInfo prefixes (RFC 8017 §9.2 note 1)
_DIGEST_INFO = {
    "sha256": bytes.fromhex("3031300d060960864801650304020105000420"),
    "sha384": bytes.fromhex("3041300d060960864801650304020205000430"),
    "sha512": bytes.fromhex("3051300d060960864801650304020305000440"),
}


def rsa_pkcs1_verify(n: int, e: int, msg: bytes, sig: bytes, hash_name: str) -> bool:
    k = (n.bit_length() + 7) // 8
    if len(sig) != k:
        return False
    s = int.from_bytes(sig, "big")
    if s >= n:
        return False
    em = pow(s, e, n).to_bytes(k, "big")
    h = hashlib.new(hash_name, msg).digest()
    t = _DIGEST_INFO[hash_name] + h
    if len(t) + 11 > k:
        return False
    expected = b"\x00\x01" + b"\xff" * (k - len(t) - 3) + b"\x00" + t
    return hmac.compare_digest(em, expected)


# ---- ECDSA (NIST P-256 / P-384) verify -------------------------------------
@dataclass(frozen=True)
class _Curve:
    p: int
    a: int
    b: int
    n: int
    gx: int
    gy: int
    size: int  # coordinate byte length
    hash_name: str


_P256 = _Curve(
    p=0xFFFFFFFF00000001000000000000000000000000FFFFFFFFFFFFFFFFFFFFFFFF,
    a=-3,
    b=0x5AC635D8AA3A93E7B3EBBD55769886BC651D06B0CC53B0F63BCE3C3E27D2604B,
    n=0xFFFFFFFF00000000FFFFFFFFFFFFFFFFBCE6FAADA7179E84F3B9CAC2FC632551,
    gx=0x6B17D1F2E12C4247F8BCE6E563A440F277037D812DEB33A0F4A13945D898C296,
    gy=0x4FE342E2FE1A7F9B8EE7EB4A7C0F9E162BCE33576B315ECECBB6406837BF51F5,
    size=32,
    hash_name="sha256",
)
_P384 = _Curve(
    p=int("fffffffffffffffffffffffffffffffffffffffffffffffffffffffffffffffeffffffff0000000000000000ffffffff", 16),
    a=-3,
    b=int("b3312fa7e23ee7e4988e056be3f82d19181d9c6efe8141120314088f5013875ac656398d8a2ed19d2a85c8edd3ec2aef", 16),
    n=int("ffffffffffffffffffffffffffffffffffffffffffffffffc7634d81f4372ddf581a0db248b0a77aecec196accc52973", 16),
    gx=int("aa87ca22be8b05378eb1c71ef320ad746e1d3b628ba79b9859f741e082542a385502f25dbf55296c3a545e3872760ab7", 16),
    gy=int("3617de4a96262c6f5d9e98bf9292dc29f8f41dbd289a147ce9da3113b5f0b8c00a60b1ce1d7e819d7a431d7c90ea0e5f", 16),
    size=48,
    hash_name="sha384",
)


def _ec_add(c: _Curve, P: Optional[Tuple[int, int]], Q: Optional[Tuple[int, int]]):
    if P is None:
        return Q
    if Q is None:
        return P
    p = c.p
    if P[0] == Q[0] and (P[1] + Q[1]) % p == 0:
        return None
    if P == Q:
        lam = (3 * P[0] * P[0] + c.a) * pow(2 * P[1], -1, p) % p
    else:
        lam = (Q[1] - P[1]) * pow(Q[0] - P[0], -1, p) % p
    x = (lam * lam - P[0] - Q[0]) % p
    y = (lam * (P[0] - x) - P[1]) % p
    return (x, y)


def _ec_mul(c: _Curve, k: int, P: Tuple[int, int]):
    R = None
    while k:
        if k & 1:
            R = _ec_add(c, R, P)
        P = _ec_add(c, P, P)
        k >>= 1
    return R


def ecdsa_verify(curve: _Curve, qx: int, qy: int, msg: bytes, sig: bytes) -> bool:
    if len(sig) != 2 * curve.size:
        return False
    r = int.from_bytes(sig[: curve.size], "big")
    s = int.from_bytes(sig[curve.size:], "big")
    if not (0 < r < curve.n and 0 < s < curve.n):
        return False
    # point on curve?
    p = curve.p
    if (qy * qy - (qx * qx * qx + curve.a * qx + curve.b)) % p != 0:
        return False
    z = int.from_bytes(hashlib.new(curve.hash_name, msg).digest(), "big")
    w = pow(s, -1, curve.n)
    u1 = z * w % curve.n
    u2 = r * w % curve.n
    R = _ec_add(c=curve, P=_ec_mul(curve, u1, (curve.gx, curve.gy)), Q=_ec_mul(curve, u2, (qx, qy)))
    return R is not None and R[0] % curve.n == r


def ecdsa_sign(curve: _Curve, d: int, msg: bytes, k: int) -> bytes:
    """Deterministic-k caller-supplied signing (used by tests and the
    control-plane token mint; k must be unique per message)."""
    z = int.from_bytes(hashlib.new(curve.hash_name, msg).digest(), "big")
    R = _ec_mul(curve, k, (curve.gx, curve.gy))
    r = R[0] % curve.n
    s = pow(k, -1, curve.n) * (z + r * d) % curve.n
    if r == 0 or s == 0:
        raise JwtError("bad ephemeral k")
    return r.to_bytes(curve.size, "big") + s.to_bytes(curve.size, "big")


# ---- JWS algorithms ---------------------------------------------------------
_HS = {"HS256": "sha256", "HS384": "sha384", "HS512": "sha512"}
_RS = {"RS256": "sha256", "RS384": "sha384", "RS512": "sha512"}
_ES = {"ES256": _P256, "ES384": _P384}


def _verify_signature(alg: str, jwk: Dict[str, Any], signing_input: bytes, sig: bytes) -> bool:
    if alg in _HS:
        key = _b64url_decode(jwk["k"])
        mac = hmac.new(key, signing_input, _HS[alg]).digest()
        return hmac.compare_digest(mac, sig)
    if alg in _RS:
        return rsa_pkcs1_verify(_b64url_to_int(jwk["n"]), _b64url_to_int(jwk["e"]), signing_input, sig, _RS[alg])
    if alg in _ES:
        c = _ES[alg]
        return ecdsa_verify(c, _b64url_to_int(jwk["x"]), _b64url_to_int(jwk["y"]), signing_input, sig)
    raise JwtError(f"unsupported algorithm {alg!r}")


def jwk_algorithm(jwk: Dict[str, Any]) -> str:
    """Key -> expected algorithm (jwt.rs jwk_to_algorithm — the
    algorithm-confusion guard compares this against the token header)."""
    if jwk.get("alg"):
        return jwk["alg"]
    kty = jwk.get("kty")
    if kty == "RSA":
        return "RS256"
    if kty == "EC":
        return {"P-256": "ES256", "P-384": "ES384"}.get(jwk.get("crv"), "ES256")
    if kty == "oct":
        return "HS256"
    raise JwtError(f"unsupported key type {kty!r}")


def encode_jwt(claims: Dict[str, Any], key: Any, alg: str, kid: Optional[str] = None) -> str:
    """Mint a JWS (control-plane tokens + tests).  `key` is bytes for HS*,
    (n, d) for RS*, (d, k) for ES* (k = unique ephemeral scalar)."""
    header: Dict[str, Any] = {"alg": alg, "typ": "JWT"}
    if kid:
        header["kid"] = kid
    signing_input = (_b64url_encode(json.dumps(header, separators=(",", ":")).encode())
                     + "." + _b64url_encode(json.dumps(claims, separators=(",", ":")).encode())).encode()
    if alg in _HS:
        sig = hmac.new(key, signing_input, _HS[alg]).digest()
    elif alg in _RS:
        n, d = key
        h = hashlib.new(_RS[alg], signing_input).digest()
        t = _DIGEST_INFO[_RS[alg]] + h
        k_len = (n.bit_length() + 7) // 8
        em = b"\x00\x01" + b"\xff" * (k_len - len(t) - 3) + b"\x00" + t
        sig = pow(int.from_bytes(em, "big"), d, n).to_bytes(k_len, "big")
    elif alg in _ES:
        d, k = key
        sig = ecdsa_sign(_ES[alg], d, signing_input, k)
    else:
        raise JwtError(f"unsupported algorithm {alg!r}")
    return signing_input.decode() + "." + _b64url_encode(sig)


# ---- JWKS provider ----------------------------------------------------------
class JwksProvider:
    """TTL-cached key set (jwks.rs:223-418).  Sources: inline dict,
    file path, or http(s) URI (aiohttp, when the deployment has egress)."""

    def __init__(self, source: Any, ttl_secs: float = 3600.0):
        self.source = source
        self.ttl_secs = ttl_secs
        self._keys: Dict[str, Dict[str, Any]] = {}
        self._fetched_at = 0.0
        if isinstance(source, dict):
            self._install(source)
            self._fetched_at = float("inf")  # inline sets never expire

    @classmethod
    async def from_issuer(cls, issuer: str, session, ttl_secs: float = 3600.0) -> "JwksProvider":
        """OIDC discovery: {issuer}/.well-known/openid-configuration -> jwks_uri."""
        url = issuer.rstrip("/") + "/.well-known/openid-configuration"
        async with session.get(url) as resp:
            if resp.status != 200:
                raise JwtError(f"OIDC discovery failed: HTTP {resp.status}")
            doc = await resp.json()
        uri = doc.get("jwks_uri")
        if not uri:
            raise JwtError("OIDC discovery document has no jwks_uri")
        return cls(uri, ttl_secs)

    def _install(self, jwks: Dict[str, Any]) -> None:
        self._keys = {k.get("kid", ""): k for k in jwks.get("keys", [])}

    async def _refresh(self, session=None) -> None:
        if isinstance(self.source, str) and self.source.startswith(("http://", "https://")):
            if session is None:
                raise JwtError("JWKS http fetch requires a client session")
            async with session.get(self.source) as resp:
                if resp.status != 200:
                    raise JwtError(f"JWKS fetch failed: HTTP {resp.status}")
                self._install(await resp.json())
        elif isinstance(self.source, str):
            with open(self.source) as f:
                self._install(json.load(f))
        self._fetched_at = time.monotonic()

    async def get_key(self, kid: str, session=None) -> Dict[str, Any]:
        if time.monotonic() - self._fetched_at > self.ttl_secs:
            await self._refresh(session)
        key = self._keys.get(kid)
        if key is None and self._fetched_at != float("inf"):
            # unknown kid -> force refresh once (rotation; jwks.rs:383-410)
            await self._refresh(session)
            key = self._keys.get(kid)
        if key is None:
            raise JwtError(f"no JWKS key with kid {kid!r}")
        return key


# ---- validator --------------------------------------------------------------
@dataclass
class ValidatedToken:
    subject: str
    issuer: str
    role: str = "user"
    email: Optional[str] = None
    name: Optional[str] = None
    claims: Dict[str, Any] = field(default_factory=dict)


class JwtValidator:
    """jwt.rs:168-509 — validate() decodes the header, fetches the kid's key,
    enforces token-alg == key-alg, verifies the signature, then checks
    exp/nbf (leeway), iss, aud, optional JTI replay, and extracts the role."""

    def __init__(
        self,
        jwks: JwksProvider,
        issuer: Optional[str] = None,
        audience: Optional[str] = None,
        leeway_secs: float = 60.0,
        role_claim: str = "roles",
        role_mapping: Optional[Dict[str, str]] = None,
        enable_jti_check: bool = False,
        jti_cache_size: int = 10_000,
        require_exp: bool = True,
    ):
        self.jwks = jwks
        self.issuer = issuer
        self.audience = audience
        self.leeway = leeway_secs
        self.role_claim = role_claim
        self.role_mapping = role_mapping or {}
        self.enable_jti_check = enable_jti_check
        self.require_exp = require_exp
        self._jti: OrderedDict[str, float] = OrderedDict()
        self._jti_cap = jti_cache_size

    async def validate(self, token: str, session=None) -> ValidatedToken:
        try:
            h_b64, c_b64, s_b64 = token.split(".")
            header = json.loads(_b64url_decode(h_b64))
            claims = json.loads(_b64url_decode(c_b64))
            sig = _b64url_decode(s_b64)
        except (ValueError, json.JSONDecodeError) as e:
            raise JwtError(f"malformed token: {e}")
        kid = header.get("kid")
        if kid is None:
            raise JwtError("token header has no kid")
        alg = header.get("alg")
        jwk = await self.jwks.get_key(kid, session)
        key_alg = jwk_algorithm(jwk)
        if alg != key_alg:  # algorithm-confusion guard (jwt.rs:277-288)
            raise JwtError(f"algorithm mismatch: token {alg!r} vs key {key_alg!r}")
        if not _verify_signature(alg, jwk, f"{h_b64}.{c_b64}".encode(), sig):
            raise JwtError("signature verification failed")
        self._check_claims(claims)
        if self.enable_jti_check and claims.get("jti"):
            self._check_jti(claims["jti"], claims.get("exp"))
        subject = claims.get("sub") or claims.get("email") or claims.get("preferred_username") or "unknown"
        return ValidatedToken(
            subject=subject,
            issuer=claims.get("iss") or (self.issuer or ""),
            role=self._extract_role(claims),
            email=claims.get("email"),
            name=claims.get("name"),
            claims=claims,
        )

    def _check_claims(self, claims: Dict[str, Any]) -> None:
        now = time.time()
        exp = claims.get("exp")
        if exp is None:
            # jsonwebtoken Validation::default() has required_spec_claims={"exp"}:
            # a token minted without exp must not validate forever
            if self.require_exp:
                raise JwtError("token has no exp claim")
        elif now > float(exp) + self.leeway:
            raise JwtError("token expired")
        nbf = claims.get("nbf")
        if nbf is not None and now < float(nbf) - self.leeway:
            raise JwtError("token not yet valid")
        if self.issuer and claims.get("iss") != self.issuer:
            raise JwtError(f"issuer mismatch: {claims.get('iss')!r}")
        if self.audience:
            aud = claims.get("aud")
            auds = aud if isinstance(aud, list) else [aud]
            if self.audience not in auds:
                raise JwtError(f"audience mismatch: {aud!r}")

    def _check_jti(self, jti: str, exp: Optional[float]) -> None:
        now = time.monotonic()
        entry = self._jti.get(jti)
        if entry is not None:
            if entry > now:
                raise JwtError(f"token replay detected (jti {jti!r})")
            del self._jti[jti]
        ttl = max(60.0, float(exp) - time.time()) if exp else 3600.0
        self._jti[jti] = now + ttl
        self._jti.move_to_end(jti)
        while len(self._jti) > self._jti_cap:
            self._jti.popitem(last=False)

    def _extract_role(self, claims: Dict[str, Any]) -> str:
        values: List[str] = []
        for name in (self.role_claim, "role", "roles", "groups", "group"):
            v = claims.get(name)
            if isinstance(v, str):
                values.append(v)
            elif isinstance(v, list):
                values.extend(x for x in v if isinstance(x, str))
            if values:
                break
        if self.role_mapping:
            for v in values:
                if v in self.role_mapping:
                    return self.role_mapping[v]
            return "user"
        for v in values:
            if v.lower() == "admin":
                return "admin"
        return "user"
