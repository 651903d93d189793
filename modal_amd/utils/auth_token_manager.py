"""Auth token manager (API-compat stub).

Parity: /root/reference/py/modal/_utils/auth_token_manager.py:18-37 — the
reference caches input-plane JWTs and refreshes them proactively at a
jittered 50–60% of token lifetime. The local control plane has no auth, but
the manager's refresh math is kept ("keep stub for API
compat") so code depending on its behavior ports cleanly.
"""

from __future__ import annotations

import random
import time
from typing import Awaitable, Callable, Optional

REFRESH_WINDOW_START = 0.5  # parity: refresh at 50-60% of lifetime
REFRESH_WINDOW_END = 0.6
RETRY_BACKOFF_S = 1.0


class AuthTokenManager:
    def __init__(self, fetch: Optional[Callable[[], Awaitable[tuple[str, float]]]] = None):
        self._fetch = fetch or self._local_fetch
        self._token: Optional[str] = None
        self._expiry: float = 0.0
        self._refresh_at: float = 0.0

    async def _local_fetch(self) -> tuple[str, float]:
        # no network, no auth: mint a dummy token with a 1h lifetime
        return f"local-{int(time.time())}", time.time() + 3600

    def _schedule_refresh(self, issued_at: float) -> None:
        lifetime = self._expiry - issued_at
        frac = random.uniform(REFRESH_WINDOW_START, REFRESH_WINDOW_END)
        self._refresh_at = issued_at + lifetime * frac

    async def get_token(self) -> str:
        now = time.time()
        if self._token is None or now >= self._refresh_at:
            try:
                self._token, self._expiry = await self._fetch()
                self._schedule_refresh(now)
            except Exception:
                if self._token is None or now >= self._expiry:
                    raise
                # keep serving the valid token; retry after backoff
                self._refresh_at = now + RETRY_BACKOFF_S
        return self._token
