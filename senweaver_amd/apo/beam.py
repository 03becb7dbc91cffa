"""Beam-search prompt optimizer — the local replacement for /api/apo/optimize.

In the reference, beam search over candidate prompts runs server-side: the
client POSTs report + rollouts + beamConfig and receives a ``beamUpdate``
{beam, bestPrompt, bestScore, round} that it folds into BeamSearchState
(apoService.ts:992-1215).  Here each round runs locally:

  1. expand: each of the ``beamWidth`` parents produces ``branchFactor``
     candidate prompts via the apply-edit decode (critique-conditioned);
  2. score: every candidate is scored against ``gradientBatchSize`` rollouts
     by the backbone — candidate-parallel across GPUs when torch.distributed
     is initialized (one-shot RCCL all-gather of the K×B score vector over
     fully-connected xGMI; see senweaver_amd/parallel/dist.py);
  3. select: Top-``beamWidth`` candidates form the next beam; historyBest is
     updated and the best prompt's '- ' rules are applied as segments.

Tie-breaking and ordering are deterministic (stable sort on (-score, version))
so every rank selects the identical beam.
"""

from __future__ import annotations

from typing import Any, Callable, Dict, List, Optional

from .optimizer import PromptOptimizerBackend
from .prompts import build_apply_edit_prompt, build_textual_gradient_prompt
from .schema import BeamSearchState, RolloutResult, VersionedPromptTemplate
from .service import APOService


class BeamSearchEngine:
    def __init__(
        self,
        backend: PromptOptimizerBackend,
        score_fn: Optional[Callable[[List[str], List[RolloutResult]], List[float]]] = None,
        clock: Optional[Callable[[], int]] = None,
        max_critique_tokens: int = 350,
        max_edit_tokens: int = 512,
        expand_fn: Optional[Callable[..., List[str]]] = None,
    ) -> None:
        """``score_fn`` scores a *batch* of candidate prompts (hook for the
        RCCL candidate-parallel path); defaults to serial backend.score.
        ``expand_fn(parent_contents, rollouts, branch_factor) -> contents``
        overrides candidate generation (e.g. rank-0-generate + broadcast)."""
        self._backend = backend
        self._score_fn = score_fn
        self._clock = clock
        self._max_critique_tokens = max_critique_tokens
        self._max_edit_tokens = max_edit_tokens
        self._expand_fn = expand_fn

    # -- candidate expansion --

    def _next_version(self, state: BeamSearchState) -> str:
        v = f"v{state.version_counter}"
        state.version_counter += 1
        return v

    def _now(self, svc: APOService) -> int:
        return self._clock() if self._clock else svc._clock()  # noqa: SLF001 — shared clock

    def expand(self, svc: APOService, state: BeamSearchState,
               rollouts: List[RolloutResult], branch_factor: int) -> List[VersionedPromptTemplate]:
        """Generate branch_factor children per beam parent via critique+edit."""
        parents = state.beam
        if not parents:
            # Round 0: seed the beam with the currently-active optimized rules
            seed_content = "\n".join(f"- {r}" for r in svc.get_optimized_rules()) or "- Be a precise, efficient coding assistant."
            parents = [VersionedPromptTemplate(
                version=self._next_version(state), content=seed_content, score=None,
                created_at=self._now(svc),
            )]
            state.beam = parents
        if self._expand_fn is not None:
            contents = self._expand_fn([p.content for p in parents], rollouts, branch_factor)
            return [VersionedPromptTemplate(
                version=self._next_version(state), content=c, score=None,
                parent_version=parents[i // branch_factor].version if i // branch_factor < len(parents) else None,
                created_at=self._now(svc),
            ) for i, c in enumerate(contents)]
        children: List[VersionedPromptTemplate] = []
        for parent in parents:
            parent_rules = [ln for ln in parent.content.split("\n") if ln.strip()]
            gradient_prompt = build_textual_gradient_prompt(parent_rules, rollouts)
            critique = self._backend.generate(gradient_prompt, max_new_tokens=self._max_critique_tokens)
            for b in range(branch_factor):
                # Vary the decode per branch by conditioning on the branch index,
                # so branches explore different edits of the same parent.
                edit_prompt = build_apply_edit_prompt(parent_rules, f"{critique}\n(Variant {b + 1}: emphasize a different single issue.)")
                content = self._backend.generate(edit_prompt, max_new_tokens=self._max_edit_tokens)
                children.append(VersionedPromptTemplate(
                    version=self._next_version(state),
                    content=content,
                    score=None,
                    parent_version=parent.version,
                    created_at=self._now(svc),
                ))
        return children

    # -- scoring --

    def score_candidates(self, candidates: List[VersionedPromptTemplate],
                         rollouts: List[RolloutResult]) -> List[float]:
        prompts = [c.content for c in candidates]
        if self._score_fn is not None:
            return list(self._score_fn(prompts, rollouts))
        return [self._backend.score(p, rollouts) for p in prompts]

    # -- one round --

    def run_round(self, svc: APOService) -> Dict[str, Any]:
        """Execute one beam round and fold the update into the service state.

        Returns the reference-shaped beamUpdate dict.
        """
        cfg = svc.get_config()
        state = svc.ensure_beam_state()
        rollouts = svc.recent_rollouts(cfg["gradientBatchSize"])
        candidates = self.expand(svc, state, rollouts, cfg["branchFactor"])
        # Parents compete with children (standard beam search keeps the best K overall)
        pool = list(state.beam) + candidates
        scores = self.score_candidates(pool, rollouts)
        for c, s in zip(pool, scores):
            c.score = s
        order = sorted(range(len(pool)), key=lambda i: (-(scores[i]), pool[i].version))
        beam_k = [pool[i] for i in order[: cfg["beamWidth"]]]
        best = beam_k[0]
        beam_update = {
            "beam": beam_k,
            "bestPrompt": best,
            "bestScore": best.score,
            "round": state.current_round + 1,
        }
        svc.apply_beam_update(beam_update)
        svc.flush()
        return beam_update

    def run_search(self, svc: APOService, rounds: Optional[int] = None) -> BeamSearchState:
        cfg = svc.get_config()
        n = rounds if rounds is not None else cfg["beamRounds"]
        for _ in range(n):
            self.run_round(svc)
        return svc.ensure_beam_state()
