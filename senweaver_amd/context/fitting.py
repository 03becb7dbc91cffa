"""4-phase context fitting — the overflow-proof message packer.

Semantics-identical to the reference's prepareMessages pipeline
(browser/convertToLLMMessageService.ts:240-500): smart output-token
reservation (20% / 16k max / 4096 min), system-message budget (30% of input,
60k-char hard cap), then
  Phase 1  aggressive message deletion past 50 messages (keep first 2 +
           last user + last 15),
  Phase 2  weighted char-level trimming to TRIM_TO_LEN=500 (weight: last
           user msg 0, assistant/tool x10, user x0.5, system x0.01,
           position ramp, first/last x0.05; <=100 iterations),
  Phase 3  emergency proportional truncation at the 85% safety margin,
  Phase 4  ultimate fallback to system + last user message.
The behavioral contract: NEVER overflow, ALWAYS answer the last user message.
"""

from __future__ import annotations

import copy
from dataclasses import dataclass
from typing import List, Optional, Tuple

CHARS_PER_TOKEN = 3.5
TRIM_TO_LEN = 500
MAX_MESSAGES_BEFORE_AGGRESSIVE_PRUNE = 50
SAFETY_MARGIN = 0.85
EMERGENCY_KEEP_CHARS = 200


@dataclass
class Msg:
    role: str  # 'system' | 'user' | 'assistant' | 'tool'
    content: str


def reserved_output_tokens(context_window: int, requested: Optional[int]) -> float:
    """Smart output reservation (reference :243-253)."""
    return max(min(context_window * 0.20, 16_000), requested if requested is not None else 4_096)


def fit_system_message(system_message: str, context_window: int,
                       reserved_output: float, ai_instructions: str = "") -> str:
    parts = []
    if ai_instructions:
        parts.append(f"GUIDELINES (from the user's .SenweaverRules file):\n{ai_instructions}")
    if system_message:
        parts.append(system_message)
    combined = "\n\n".join(parts)
    available_chars = (context_window - reserved_output) * CHARS_PER_TOKEN
    budget = min(available_chars * 0.30, 60_000)
    if len(combined) > budget:
        combined = combined[: int(budget) - 40] + "\n...[system prompt truncated for context budget]..."
    return combined


def _last_user_idx(messages: List[Msg]) -> int:
    for i in range(len(messages) - 1, -1, -1):
        if messages[i].role == "user":
            return i
    return -1


def prepare_messages(messages_in: List[Msg], system_message: str, context_window: int,
                     reserved_output_token_space: Optional[int] = None,
                     ai_instructions: str = "") -> Tuple[str, List[Msg]]:
    """Returns (final_system_message, fitted_messages)."""
    reserved = reserved_output_tokens(context_window, reserved_output_token_space)
    combined_sys = fit_system_message(system_message, context_window, reserved, ai_instructions)
    messages: List[Msg] = [Msg("system", combined_sys)] + [copy.copy(m) for m in messages_in]
    for m in messages:
        if m.role != "tool":
            m.content = m.content.strip()
    last_user = _last_user_idx(messages)

    # ---- Phase 1: aggressive deletion ----
    if len(messages) > MAX_MESSAGES_BEFORE_AGGRESSIVE_PRUNE:
        keep = set(range(min(2, len(messages))))
        if last_user >= 0:
            keep.add(last_user)
        keep.update(range(max(0, len(messages) - 15), len(messages)))
        messages = [m for i, m in enumerate(messages) if i in keep]
        last_user = _last_user_idx(messages)

    # ---- Phase 2: weighted char-level trimming ----
    already_trimmed = set()

    def weight(idx: int) -> float:
        if idx == last_user:
            return 0.0
        m = messages[idx]
        base = len(m.content)
        mult = 1 + (len(messages) - 1 - idx) / len(messages)
        if m.role == "user":
            mult *= 0.5
        elif m.role == "system":
            mult *= 0.01
        else:
            mult *= 10
        if idx in already_trimmed:
            mult = 0
        if idx <= 1 or idx >= len(messages) - 1 - 3:
            mult *= 0.05
        return base * mult

    total_len = sum(len(m.content) for m in messages)
    available_input_chars = (context_window - reserved) * CHARS_PER_TOKEN
    chars_to_trim = total_len - max(available_input_chars, 20_000)
    if chars_to_trim > 0:
        remaining = chars_to_trim
        for _ in range(100):
            if remaining <= 0:
                break
            idx = max(range(len(messages)), key=weight, default=-1)
            if idx == -1:
                break
            if idx == last_user:
                already_trimmed.add(idx)
                continue
            m = messages[idx]
            if len(m.content) <= TRIM_TO_LEN:
                already_trimmed.add(idx)
                if len(already_trimmed) >= len(messages) - 3:
                    if len(messages) > 10:
                        keep = set(range(min(2, len(messages))))
                        if last_user >= 0:
                            keep.add(last_user)
                        keep.update(range(max(0, len(messages) - 3), len(messages)))
                        messages = [mm for i, mm in enumerate(messages) if i in keep]
                        last_user = _last_user_idx(messages)
                    break
                continue
            will_trim = len(m.content) - TRIM_TO_LEN
            if will_trim > remaining:
                m.content = m.content[: len(m.content) - int(remaining) - 3].strip() + "..."
                break
            remaining -= will_trim
            m.content = m.content[: TRIM_TO_LEN - 3] + "..."
            already_trimmed.add(idx)

    # ---- Phase 3: emergency proportional truncation ----
    final_len = sum(len(m.content) for m in messages)
    safe_chars = available_input_chars * SAFETY_MARGIN
    if final_len > safe_chars:
        ratio = safe_chars / final_len
        for idx in range(1, len(messages)):
            m = messages[idx]
            if m.role == "system" or idx == last_user:
                continue
            target = max(EMERGENCY_KEEP_CHARS, int(len(m.content) * ratio))
            if len(m.content) > target:
                m.content = m.content[: target - 30] + "\n...[emergency truncation]..."
        recheck = sum(len(m.content) for m in messages)
        if recheck > safe_chars and len(messages) > 4:
            keep = {0}
            if last_user >= 0:
                keep.add(last_user)
            keep.update(range(max(0, len(messages) - 3), len(messages)))
            messages = [m for i, m in enumerate(messages) if i in keep]
            last_user = _last_user_idx(messages)

    # ---- Phase 4: ultimate fallback ----
    ultimate = sum(len(m.content) for m in messages)
    if ultimate > available_input_chars:
        sys_msg = next((m for m in messages if m.role == "system"), None)
        last_msg = messages[last_user] if last_user >= 0 else messages[-1]
        messages = []
        if sys_msg is not None:
            max_sys = max(2000, int(available_input_chars - len(last_msg.content) - 1000))
            if len(sys_msg.content) > max_sys:
                sys_msg.content = sys_msg.content[: max_sys - 30] + "\n...[system message truncated]..."
            messages.append(sys_msg)
        messages.append(last_msg)

    final_system = messages.pop(0).content if messages and messages[0].role == "system" else combined_sys
    return final_system, messages
