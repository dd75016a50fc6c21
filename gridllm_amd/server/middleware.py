"""Gateway rate limiting.

The reference shipped a fixed-window in-memory limiter with X-RateLimit
headers but never mounted it (client/src/middleware/rateLimiter.ts:63-130,
unmounted per SURVEY.md §2.2 C9). Here it is mounted on the gateway,
disabled by default (limit 0) and enabled via RATE_LIMIT_MAX_REQUESTS /
RATE_LIMIT_WINDOW_MS, matching the reference's env knob names.
"""

from __future__ import annotations

import time

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse


class FixedWindowLimiter:
    def __init__(self, max_requests: int, window_ms: int):
        self.max_requests = max_requests
        self.window_ms = window_ms
        self._windows: dict[str, tuple[int, int]] = {}  # key -> (start_ms, n)

    def check(self, key: str, now_ms: int | None = None
              ) -> tuple[bool, int, int]:
        """Returns (allowed, remaining, reset_ms)."""
        now = now_ms if now_ms is not None else int(time.time() * 1000)
        start, n = self._windows.get(key, (now, 0))
        if now - start >= self.window_ms:
            start, n = now, 0
        n += 1
        self._windows[key] = (start, n)
        reset = start + self.window_ms
        remaining = max(0, self.max_requests - n)
        if len(self._windows) > 16384:      # bound memory: drop dead windows
            self._windows = {k: v for k, v in self._windows.items()
                             if now - v[0] < self.window_ms}
        return n <= self.max_requests, remaining, reset


def mount_rate_limiter(app: FastAPI, max_requests: int,
                       window_ms: int) -> None:
    """No-op when max_requests <= 0 (default: unlimited)."""
    if max_requests <= 0:
        return
    limiter = FixedWindowLimiter(max_requests, window_ms)

    @app.middleware("http")
    async def rate_limit(request: Request, call_next):
        if request.url.path.startswith("/health"):
            return await call_next(request)
        client = request.client.host if request.client else "unknown"
        allowed, remaining, reset = limiter.check(client)
        if not allowed:
            return JSONResponse(
                status_code=429,
                content={"error": {"message": "Too many requests",
                                   "code": "RATE_LIMITED"}},
                headers={
                    "X-RateLimit-Limit": str(max_requests),
                    "X-RateLimit-Remaining": "0",
                    "X-RateLimit-Reset": str(reset),
                    "Retry-After": str(max(1, (reset - int(time.time() * 1000)) // 1000)),
                })
        resp = await call_next(request)
        resp.headers["X-RateLimit-Limit"] = str(max_requests)
        resp.headers["X-RateLimit-Remaining"] = str(remaining)
        resp.headers["X-RateLimit-Reset"] = str(reset)
        return resp
