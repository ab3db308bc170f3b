"""HTTP transport seam for the live connectors.

Every live client (Binance REST, LunarCrush, news feeds) talks through a
`Transport`: `request(method, url, params, headers, data) -> (status,
body_text)`. Three implementations:

  - UrllibTransport: real network via stdlib urllib (no extra deps);
    what production uses.
  - RecordingTransport: wraps another transport and appends every
    exchange to a JSONL tape — how fixture files are produced in an
    environment with egress.
  - ReplayTransport: serves a recorded JSONL tape; requests are matched
    by (method, path, filtered params) in order with fallback to
    any-order matching. This is what the offline test suite uses (this
    container has no network), mirroring the reference's live calls
    (services/utils/exchange_interface.py:67-207,
    backtesting/data_manager.py:47-114) without credentials.

Secrets (API keys, signatures, timestamps) are stripped before matching
so tapes stay replayable and contain no credentials.
"""

from __future__ import annotations

import json
import urllib.error
import urllib.parse
import urllib.request
from pathlib import Path

# volatile params never used for replay matching
_VOLATILE = {"timestamp", "signature", "recvWindow", "listenKey"}


class TransportError(Exception):
    pass


class UrllibTransport:
    def __init__(self, timeout: float = 10.0):
        self.timeout = timeout

    def request(self, method: str, url: str, params: dict | None = None,
                headers: dict | None = None,
                data: bytes | None = None) -> tuple[int, str]:
        if params:
            url = url + "?" + urllib.parse.urlencode(params)
        req = urllib.request.Request(url, data=data, method=method,
                                     headers=headers or {})
        try:
            with urllib.request.urlopen(req, timeout=self.timeout) as r:
                return r.status, r.read().decode()
        except urllib.error.HTTPError as e:
            return e.code, e.read().decode()


def _match_key(method: str, url: str, params: dict | None):
    path = urllib.parse.urlparse(url).path
    p = {k: str(v) for k, v in (params or {}).items()
         if k not in _VOLATILE}
    return method.upper(), path, tuple(sorted(p.items()))


class RecordingTransport:
    def __init__(self, inner, tape_path: str | Path):
        self.inner = inner
        self.tape_path = Path(tape_path)

    def request(self, method, url, params=None, headers=None, data=None):
        status, body = self.inner.request(method, url, params, headers,
                                          data)
        m, path, p = _match_key(method, url, params)
        with open(self.tape_path, "a") as f:
            f.write(json.dumps({
                "method": m, "path": path, "params": dict(p),
                "status": status, "body": body,
            }) + "\n")
        return status, body


class ReplayTransport:
    """Replays a JSONL tape. Entries matching (method, path, params) are
    consumed in recorded order; repeated identical requests re-serve the
    last matching entry once the tape position passes it (steady-state
    endpoints like tickers are usually recorded once)."""

    def __init__(self, tape_path: str | Path | None = None,
                 entries: list[dict] | None = None, strict: bool = True):
        if entries is None:
            entries = [json.loads(ln) for ln in
                       Path(tape_path).read_text().splitlines() if ln]
        self.entries = entries
        self.consumed = [False] * len(entries)
        self.strict = strict
        self.unmatched: list[tuple] = []

    def request(self, method, url, params=None, headers=None, data=None):
        m, path, p = _match_key(method, url, params)
        want = dict(p)
        # first unconsumed exact match, else last consumed exact match
        fallback = None
        for i, e in enumerate(self.entries):
            if e["method"] == m and e["path"] == path \
                    and e.get("params", {}) == want:
                if not self.consumed[i]:
                    self.consumed[i] = True
                    return e["status"], e["body"]
                fallback = e
        if fallback is not None:
            return fallback["status"], fallback["body"]
        # relaxed: match on path only (params differ e.g. by paging)
        for i, e in enumerate(self.entries):
            if e["method"] == m and e["path"] == path \
                    and not self.consumed[i]:
                self.consumed[i] = True
                return e["status"], e["body"]
        self.unmatched.append((m, path, want))
        if self.strict:
            raise TransportError(f"no tape entry for {m} {path} {want}")
        return 404, "{}"
