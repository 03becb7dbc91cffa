from .fitting import (
    CHARS_PER_TOKEN,
    TRIM_TO_LEN,
    Msg,
    fit_system_message,
    prepare_messages,
    reserved_output_tokens,
)
from .compress import (
    KEEP_RECENT_COUNT,
    MODEL_CONTEXT_LIMITS,
    SMART_CONTEXT_CONFIG,
    CompressibleMessage,
    EnhancedContextManager,
    TokenEstimator,
    compact_tool_output,
    compress_old_messages,
    needs_compression,
)
from .pipeline import ConvertToLLMMessages, base_system_message, multi_agent_section
