from .limits import (
    CHAT_RETRIES,
    TPMRateLimiter,
    get_retry_delay_ms,
    is_context_length_error,
    is_rate_limit_error,
)
from .thread import ChatThread, ChatThreadService, GlobalSettings, ThreadMessage
