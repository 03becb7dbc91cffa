from .grammar import (
    PARAM_ALIASES,
    RawToolCall,
    ReasoningExtractor,
    XMLToolExtractor,
    ends_with_any_prefix_of,
    parse_xml_prefix_to_tool_call,
    trim_before_and_after_newlines,
)
from .service import LLMChatMessage, LLMMessageService

__all__ = ["PARAM_ALIASES", "RawToolCall", "ReasoningExtractor", "XMLToolExtractor",
           "ends_with_any_prefix_of", "parse_xml_prefix_to_tool_call",
           "trim_before_and_after_newlines", "LLMChatMessage", "LLMMessageService"]
