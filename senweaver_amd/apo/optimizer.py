"""Optimizer backend protocol — the seam between APO logic and the GPU scorer.

The reference delegated critique generation and candidate scoring to its HTTPS
backend (apoService.ts:992-1343).  Here the same two capabilities are a local
protocol, implemented by:

- ``senweaver_amd.engine.scorer.LlamaBackend`` — the MI355X Llama backbone
  (hand-written HIP kernels), optionally candidate-parallel over RCCL/xGMI;
- ``StubBackend`` below — a deterministic CPU stand-in for unit tests.
"""

from __future__ import annotations

import hashlib
from typing import List, Optional, Protocol

from .schema import RolloutResult


class PromptOptimizerBackend(Protocol):
    """What the APO engines need from a model backend."""

    def generate(self, prompt: str, max_new_tokens: int = 256) -> str:
        """Greedy-decode a completion for ``prompt``."""
        ...

    def score(self, candidate_prompt: str, rollouts: List[RolloutResult]) -> float:
        """Score a candidate system prompt against rollout conversations.

        Contract: higher = better.  The GPU backend computes a reward-weighted
        mean assistant-token log-likelihood: for each rollout r with reward
        w_r = finalReward (or ±1 from status when reward is null), score =
        sum_r w_r * avg_logprob(assistant tokens | candidate + context_r) / sum_r |w_r|.
        Candidates that make high-reward conversations more likely and
        low-reward ones less likely score higher.
        """
        ...


class StubBackend:
    """Deterministic CPU backend for tests: hash-based 'generation' and scoring."""

    def __init__(self, seed: int = 0) -> None:
        self.seed = seed
        self.generate_calls: List[str] = []
        self.score_calls: List[str] = []

    def _h(self, text: str) -> int:
        return int.from_bytes(hashlib.sha256(f"{self.seed}:{text}".encode()).digest()[:8], "big")

    def generate(self, prompt: str, max_new_tokens: int = 256) -> str:
        self.generate_calls.append(prompt)
        h = self._h(prompt)
        lines = [f"- Rule variant {h % 1000}: respond concisely and verify tool output.",
                 f"- Rule variant {(h >> 16) % 1000}: prefer minimal tool calls."]
        return "\n".join(lines)

    def score(self, candidate_prompt: str, rollouts: List[RolloutResult]) -> float:
        self.score_calls.append(candidate_prompt)
        return (self._h(candidate_prompt) % 10_000) / 10_000.0


def rollout_weight(r: RolloutResult) -> float:
    """Reward weight for scoring: finalReward when present, else ±1 from status."""
    if r.final_reward is not None:
        return r.final_reward
    if r.status == "succeeded":
        return 1.0
    if r.status == "failed":
        return -1.0
    return 0.0


class LocalGradientEngine:
    """Local textual-gradient service (replaces POST {api}/apo/gradient).

    Runs the reference's exact gradient prompt through the backend to get the
    critique, then substitutes the critique into the apply-edit prompt and
    decodes the improved rules.  The response shape matches the reference's
    backend contract: {"critique": str, "editedPrompt": str}.
    """

    CRITIQUE_PLACEHOLDER = "{{critique_placeholder}}"

    def __init__(self, backend: PromptOptimizerBackend, max_critique_tokens: int = 350,
                 max_edit_tokens: int = 512) -> None:
        self._backend = backend
        self._max_critique_tokens = max_critique_tokens
        self._max_edit_tokens = max_edit_tokens

    def textual_gradient(self, gradient_prompt: str, apply_edit_prompt: str,
                         rollout_results: List[RolloutResult],
                         current_rules: List[str]) -> Optional[dict]:
        critique = self._backend.generate(gradient_prompt, max_new_tokens=self._max_critique_tokens)
        if not critique:
            return None
        edit_prompt = apply_edit_prompt.replace(self.CRITIQUE_PLACEHOLDER, critique)
        edited = self._backend.generate(edit_prompt, max_new_tokens=self._max_edit_tokens)
        # Keep only '- ' rule lines, the format the reference's apply path expects
        rule_lines = [ln for ln in edited.split("\n") if ln.strip().startswith("- ")]
        edited_prompt = "\n".join(rule_lines) if rule_lines else edited
        return {"critique": critique, "editedPrompt": edited_prompt}
