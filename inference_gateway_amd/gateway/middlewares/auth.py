"""OIDC bearer-token authentication middleware.

Parity with the reference (api/middlewares/auth.go:32-87): OIDC discovery
against AUTH_OIDC_ISSUER at startup, bearer JWT verification on every
request, verified claims stashed in request state. JWT verification
(RS256 via JWKS, HS256 via the client secret) is implemented directly —
RSASSA-PKCS1-v1_5 verification is one modular exponentiation plus a digest
comparison, so no crypto dependency is needed.
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Dict, Optional

from ..http import Request, Response


def _b64url_decode(s: str) -> bytes:
    s = s + "=" * (-len(s) % 4)
    return base64.urlsafe_b64decode(s)


def _b64url_to_int(s: str) -> int:
    return int.from_bytes(_b64url_decode(s), "big")


# DigestInfo prefix for SHA-256 (RFC 8017 §9.2)
_SHA256_PREFIX = bytes.fromhex("3031300d060960864801650304020105000420")


def verify_rs256(n: int, e: int, signing_input: bytes, sig: bytes) -> bool:
    k = (n.bit_length() + 7) // 8
    if len(sig) != k:
        return False
    em = pow(int.from_bytes(sig, "big"), e, n).to_bytes(k, "big")
    digest = hashlib.sha256(signing_input).digest()
    expected = (b"\x00\x01" + b"\xff" * (k - len(digest)
                                         - len(_SHA256_PREFIX) - 3)
                + b"\x00" + _SHA256_PREFIX + digest)
    return hmac.compare_digest(em, expected)


class OIDCAuthenticator:
    def __init__(self, auth_cfg, client, logger):
        self.cfg = auth_cfg
        self.client = client
        self.logger = logger
        self.jwks: Dict[str, dict] = {}   # kid -> jwk
        self.issuer = auth_cfg.oidc_issuer.rstrip("/")

    async def initialize(self) -> None:
        disc = await self.client.get(
            self.issuer + "/.well-known/openid-configuration")
        disc.raise_for_status()
        jwks_uri = disc.json()["jwks_uri"]
        jwks = await self.client.get(jwks_uri)
        jwks.raise_for_status()
        for key in jwks.json().get("keys", []):
            if key.get("kid"):
                self.jwks[key["kid"]] = key
        self.logger.info("OIDC initialized", "issuer", self.issuer,
                         "keys", len(self.jwks))

    # ------------------------------------------------------------------
    def verify(self, token: str) -> Optional[dict]:
        """Returns claims dict or None."""
        try:
            h_b64, p_b64, s_b64 = token.split(".")
            header = json.loads(_b64url_decode(h_b64))
            claims = json.loads(_b64url_decode(p_b64))
            sig = _b64url_decode(s_b64)
            signing_input = (h_b64 + "." + p_b64).encode()
        except Exception:
            return None
        alg = header.get("alg")
        if alg == "RS256":
            jwk = self.jwks.get(header.get("kid"))
            if jwk is None and len(self.jwks) == 1:
                jwk = next(iter(self.jwks.values()))
            if jwk is None:
                return None
            if not verify_rs256(_b64url_to_int(jwk["n"]),
                                _b64url_to_int(jwk["e"]),
                                signing_input, sig):
                return None
        elif alg == "HS256":
            secret = self.cfg.oidc_client_secret.encode()
            if not secret:
                # public client (no secret): HS256 would be verifiable
                # with an EMPTY key — an attacker-forgeable token
                return None
            expected = hmac.new(secret, signing_input,
                                hashlib.sha256).digest()
            if not hmac.compare_digest(expected, sig):
                return None
        else:
            return None
        now = time.time()
        # exp is REQUIRED (go-oidc semantics — a token that never
        # expires is rejected, not accepted forever)
        if "exp" not in claims or now > float(claims["exp"]):
            return None
        if "nbf" in claims and now < float(claims["nbf"]):
            return None
        iss = claims.get("iss", "").rstrip("/")
        if iss and iss != self.issuer:
            return None
        # audience check (go-oidc Config.ClientID): aud must name this
        # client (string or array form)
        cid = self.cfg.oidc_client_id
        if cid:
            aud = claims.get("aud")
            auds = aud if isinstance(aud, list) else [aud]
            if cid not in auds:
                return None
        return claims

    def middleware(self):
        async def mw(req: Request, nxt) -> Response:
            if req.path == "/health":
                return await nxt(req)
            authz = req.headers.get("authorization", "")
            if not authz.lower().startswith("bearer "):
                return Response.error(401, "missing bearer token")
            token = authz[7:].strip()
            claims = self.verify(token)
            if claims is None:
                return Response.error(401, "invalid token")
            req.state["auth_claims"] = claims
            req.state["auth_token"] = token
            return await nxt(req)

        return mw
