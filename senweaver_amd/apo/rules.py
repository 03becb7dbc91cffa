"""2000-character APO rule injection budget packer.

Semantics-identical to the injection block in the reference's system-message
assembly (browser/convertToLLMMessageService.ts:832-853): greedy packing of
rules joined by '\n' under APO_RULES_MAX_CHARS, with the
"# APO Optimized Rules (k/n rules, budget limited)" header when truncated.
"""

from __future__ import annotations

from typing import List, Tuple

from .schema import APO_RULES_MAX_CHARS


def pack_rules(rules: List[str], max_chars: int = APO_RULES_MAX_CHARS) -> Tuple[str, int]:
    """Greedy-pack rules under the char budget.

    Returns (packed_content, rules_included).  Identical to the reference loop:
    the first rule that would push the joined content over ``max_chars`` stops
    packing (even if a later, shorter rule would fit).
    """
    content = ""
    included = 0
    for rule in rules:
        candidate = content + ("\n" if content else "") + rule
        if len(candidate) > max_chars:
            break
        content = candidate
        included += 1
    return content, included


def inject_rules(system_message: str, rules: List[str], max_chars: int = APO_RULES_MAX_CHARS) -> str:
    """Append the APO-optimized-rules section to a system message."""
    if not rules:
        return system_message
    try:
        content, included = pack_rules(rules, max_chars)
        if content:
            trunc_note = f" ({included}/{len(rules)} rules, budget limited)" if included < len(rules) else ""
            system_message += f"\n\n# APO Optimized Rules{trunc_note}\n" + content
    except Exception:
        pass  # APO failure never breaks message assembly (reference :851-853)
    return system_message
