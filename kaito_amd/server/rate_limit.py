"""Queue-depth 429 rate limiting middleware.

Parity with the reference's Starlette middleware
(presets/workspace/inference/vllm/rate_limit.py): reject generation
requests with 429 when the engine's waiting queue exceeds max_num_seqs;
non-generation endpoints (health/metrics) are never guarded.
"""
from __future__ import annotations

from starlette.middleware.base import BaseHTTPMiddleware
from starlette.requests import Request
from starlette.responses import JSONResponse

from . import metrics

GUARDED_PREFIXES = ("/v1/completions", "/v1/chat/completions", "/v1/embeddings")


class RateLimitMiddleware(BaseHTTPMiddleware):
    def __init__(self, app, get_queue_depth, max_queue: int):
        super().__init__(app)
        self.get_queue_depth = get_queue_depth
        self.max_queue = max_queue

    async def dispatch(self, request: Request, call_next):
        path = request.url.path
        if any(path.startswith(p) for p in GUARDED_PREFIXES):
            if self.get_queue_depth() >= self.max_queue:
                metrics.RATELIMIT_REJECTED.inc()
                return JSONResponse(
                    {"error": {
                        "message": "server overloaded: request queue full",
                        "type": "rate_limit_exceeded", "code": 429}},
                    status_code=429,
                    headers={"Retry-After": "1"})
        return await call_next(request)
