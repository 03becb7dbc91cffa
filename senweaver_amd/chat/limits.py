"""Reactive rate limiting + retry ladder constants.

Rebuilds the reference's TPMRateLimiter (common/tpmRateLimiter.ts):
REACTIVE limiting — no predictive pre-wait; a 429 sets a cooldown (with
retry-after extraction) and getWaitTime gates the next request; success
clears the cooldown.  Retry constants from browser/chatThreadService.ts:52-63.
"""

from __future__ import annotations

import math
import re
import time
from typing import Dict

CHAT_RETRIES = 5
BASE_RETRY_DELAY_MS = 3000
MAX_RETRY_DELAY_MS = 60000

DEFAULT_TPM_CONFIGS: Dict[str, dict] = {
    # the local backbone has no provider quota; limits exist for the
    # remote-provider-compatible surface and tests
    "local": {"tokensPerMinute": math.inf, "requestsPerMinute": math.inf,
              "enablePredictiveRateLimiting": False, "minRequestInterval": 0},
    "default": {"tokensPerMinute": 200000, "requestsPerMinute": 500,
                "enablePredictiveRateLimiting": False, "minRequestInterval": 100},
}


def get_retry_delay_ms(attempt: int, is_tpm_error: bool) -> float:
    if is_tpm_error:
        return min(BASE_RETRY_DELAY_MS * 2 ** attempt, MAX_RETRY_DELAY_MS)
    return min(BASE_RETRY_DELAY_MS * 1.5 ** (attempt - 1), MAX_RETRY_DELAY_MS / 2)


_CONTEXT_ERR_PATTERNS = ("context_length", "context length", "maximum context",
                         "token limit", "too many tokens", "max_tokens",
                         "input is too long")


def is_context_length_error(error_text: str) -> bool:
    """Error-string matching from chatThreadService.ts:1437-1446."""
    s = error_text.lower()
    if any(p in s for p in _CONTEXT_ERR_PATTERNS):
        return True
    return "400" in s and ("token" in s or "length" in s)


def is_rate_limit_error(error_text: str) -> bool:
    s = error_text.lower()
    return "429" in s or "rate limit" in s or "rate_limit" in s or "quota" in s


_RETRY_AFTER_RE = re.compile(r"retry[-_]after[\"':\s]*([0-9.]+)", re.I)


class TPMRateLimiter:
    def __init__(self, clock=None) -> None:
        self._clock = clock or (lambda: time.time() * 1000)
        self._cooldown_until: Dict[str, float] = {}
        self._last_request: Dict[str, float] = {}

    def _config(self, provider: str) -> dict:
        return DEFAULT_TPM_CONFIGS.get(provider, DEFAULT_TPM_CONFIGS["default"])

    def get_wait_time_ms(self, provider: str) -> float:
        now = self._clock()
        wait = max(0.0, self._cooldown_until.get(provider, 0) - now)
        min_iv = self._config(provider)["minRequestInterval"]
        last = self._last_request.get(provider)
        if last is not None and min_iv:
            wait = max(wait, min_iv - (now - last))
        return wait

    def record_request(self, provider: str) -> None:
        self._last_request[provider] = self._clock()

    def handle_rate_limit_error(self, provider: str, error_text: str,
                                attempt: int = 0) -> float:
        """Returns the cooldown in ms (retry-after from the error when present)."""
        m = _RETRY_AFTER_RE.search(error_text)
        if m:
            cooldown = float(m.group(1)) * 1000
        else:
            cooldown = get_retry_delay_ms(attempt, is_tpm_error=True)
        self._cooldown_until[provider] = self._clock() + cooldown
        return cooldown

    def record_success(self, provider: str) -> None:
        self._cooldown_until.pop(provider, None)
