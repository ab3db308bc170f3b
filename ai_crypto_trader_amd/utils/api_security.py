"""API security manager (reference parity: services/utils/api_security.py
:60-693 — key issue/rotate/revoke with hashing, access levels, IP
whitelist, audit log, 30-day rotation scheduler)."""

from __future__ import annotations

import hashlib
import hmac
import json
import secrets
import time
from enum import Enum
from pathlib import Path


class AccessLevel(str, Enum):
    READ_ONLY = "read_only"
    TRADE = "trade"
    ADMIN = "admin"


LEVEL_RANK = {AccessLevel.READ_ONLY: 0, AccessLevel.TRADE: 1,
              AccessLevel.ADMIN: 2}
ROTATION_DAYS = 30


class APISecurityManager:
    def __init__(self, store_path: str | None = None):
        self.keys: dict[str, dict] = {}       # key_id -> record
        self.audit: list[dict] = []
        self.store_path = Path(store_path) if store_path else None
        if self.store_path and self.store_path.exists():
            self.keys = json.loads(self.store_path.read_text())

    # --- key lifecycle ---------------------------------------------------
    @staticmethod
    def _hash(secret: str) -> str:
        return hashlib.sha256(secret.encode()).hexdigest()

    def issue_key(self, owner: str,
                  level: AccessLevel = AccessLevel.READ_ONLY,
                  ip_whitelist: list[str] | None = None) -> tuple[str, str]:
        """Returns (key_id, secret) — the secret is stored hashed only."""
        key_id = secrets.token_hex(8)
        secret = secrets.token_urlsafe(32)
        self.keys[key_id] = {
            "owner": owner, "level": level.value,
            "hash": self._hash(secret),
            "ip_whitelist": ip_whitelist or [],
            "issued_at": time.time(), "active": True,
        }
        self._log("issue", key_id, owner)
        self._persist()
        return key_id, secret

    def rotate_key(self, key_id: str) -> str | None:
        rec = self.keys.get(key_id)
        if not rec or not rec["active"]:
            return None
        secret = secrets.token_urlsafe(32)
        rec["hash"] = self._hash(secret)
        rec["issued_at"] = time.time()
        self._log("rotate", key_id, rec["owner"])
        self._persist()
        return secret

    def revoke_key(self, key_id: str) -> bool:
        rec = self.keys.get(key_id)
        if not rec:
            return False
        rec["active"] = False
        self._log("revoke", key_id, rec["owner"])
        self._persist()
        return True

    # --- auth ------------------------------------------------------------
    def authenticate(self, key_id: str, secret: str,
                     required: AccessLevel = AccessLevel.READ_ONLY,
                     ip: str | None = None) -> bool:
        rec = self.keys.get(key_id)
        ok = bool(
            rec and rec["active"]
            and hmac.compare_digest(rec["hash"], self._hash(secret))
            and LEVEL_RANK[AccessLevel(rec["level"])]
            >= LEVEL_RANK[required]
            and (not rec["ip_whitelist"] or ip in rec["ip_whitelist"])
        )
        self._log("auth_ok" if ok else "auth_fail", key_id,
                  rec["owner"] if rec else "?", ip=ip)
        return ok

    def keys_needing_rotation(self, now: float | None = None) -> list[str]:
        now = now or time.time()
        horizon = ROTATION_DAYS * 86400
        return [k for k, r in self.keys.items()
                if r["active"] and now - r["issued_at"] > horizon]

    # --- audit -----------------------------------------------------------
    def _log(self, action: str, key_id: str, owner: str, **kw):
        self.audit.append({"at": time.time(), "action": action,
                           "key_id": key_id, "owner": owner, **kw})
        del self.audit[:-10_000]

    def _persist(self):
        if self.store_path:
            self.store_path.parent.mkdir(parents=True, exist_ok=True)
            self.store_path.write_text(json.dumps(self.keys, indent=2))
