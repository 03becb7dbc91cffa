"""Textual-gradient / apply-edit prompt templates.

Text-identical to the reference's _buildTextualGradientPrompt
(apoService.ts:918-962) and _buildApplyEditPrompt (:966-988) so a critique
model fine-tuned or evaluated against the reference's prompts behaves the same.
"""

from __future__ import annotations

from typing import List

from ..utils.jsonutil import to_fixed
from .schema import RolloutResult


def format_rollout_experiments(rollout_results: List[RolloutResult]) -> str:
    sections = []
    for i, r in enumerate(rollout_results):
        status_text = "✅ Succeeded" if r.status == "succeeded" else "❌ Failed" if r.status == "failed" else "❓ Unknown"
        reward_text = to_fixed(r.final_reward, 3) if r.final_reward is not None else "N/A"
        msg_summary = "\n    ".join(f"[{m.role}] {m.content[:200]}" for m in r.messages)
        tc = r.tool_call_stats
        if tc["totalCalls"] > 0:
            rate = f"{to_fixed(tc['successRate'] * 100, 0)}%" if tc["successRate"] is not None else "N/A"
            tool_info = (
                f"Tool Calls: {tc['totalCalls']} ({tc['succeeded']} succeeded, {tc['failed']} failed, "
                f"rate: {rate}, duration: {to_fixed(tc['totalDurationMs'], 0)}ms)"
            )
        else:
            tool_info = "Tool Calls: none"
        if r.reward_dimensions:
            reward_dims = "Reward Dims: " + ", ".join(
                f"{d['name']}={to_fixed(d['value'], 2)}" for d in r.reward_dimensions)
        else:
            reward_dims = ""
        llm_info = f"LLM Calls: {r.llm_stats['totalCalls']}, Tokens: {r.llm_stats['totalTokens']}"
        sections.append(
            f"--- Experiment {i + 1} ---\nStatus: {status_text}\nFinal Reward: {reward_text}\n"
            f"Chat Mode: {r.chat_mode}\n{tool_info}\n{llm_info}\n{reward_dims}\nMessages:\n    {msg_summary}"
        )
    return "\n\n".join(sections)


def build_textual_gradient_prompt(current_prompt_rules: List[str], rollout_results: List[RolloutResult]) -> str:
    prompt_section = "\n".join(current_prompt_rules) if current_prompt_rules else "(No optimized prompt rules currently active)"
    experiments_section = format_rollout_experiments(rollout_results)
    return f"""You are an expert prompt engineer optimizing a coding IDE assistant's system prompt.

## Current Prompt Rules
{prompt_section}

## Sample Runs with Current Prompt
{experiments_section}

## Your Task
Produce a brief critique listing specific causes for failures or ways to raise reward next time.
Return a bullet list with concrete, testable changes (format, constraints, ordering, definitions).
Focus on:
1. Structural issues: missing goals, contradictions, no stop conditions
2. Instruction quality: vague verbs, lack of hierarchy, overlapping scope
3. Control and behavior: tool limits, uncertainty handling, verbosity
4. Input/output specification: missing defaults, format inconsistency
5. Scope and safety: scope creep, unsafe actions, error handling

Be concise and direct. Less than 350 words."""


def build_apply_edit_prompt(current_prompt_rules: List[str], critique: str) -> str:
    prompt_section = "\n".join(current_prompt_rules) if current_prompt_rules else "(No optimized prompt rules currently active)"
    return f"""Revise the given prompt rules using the critique as constraints and improvement guide.

## Revision Rules
1. Rewrite or restructure the prompt if critique implies it.
2. Explicitly include any requested output format, structure, or word limit.
3. Prioritize mechanism-first phrasing: define what to do, then how to do it.
4. Keep the new prompt close in tone, length, and structure to the original.
5. Focus on the single most critical issue from the critique.

## Current Prompt Rules
{prompt_section}

## Critique
{critique}

Return only the improved prompt rules. Do not include explanations or headers.
Each rule should be on its own line, starting with "- "."""
