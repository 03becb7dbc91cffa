"""LlamaBackend — the local optimizer backend (generation + candidate scoring).

Implements ``senweaver_amd.apo.optimizer.PromptOptimizerBackend`` on the
MI355X Llama backbone: greedy decode for textual-gradient critiques and
apply-edit rewrites, and reward-weighted teacher-forced log-likelihood for
beam-candidate scoring (the work the reference shipped to its backend via
POST /api/apo/gradient and /api/apo/optimize).

Scoring definition (see optimizer.PromptOptimizerBackend.score):
    score(c) = sum_r w_r * avg_logprob(assistant tokens | c + context_r)
               / sum_r |w_r|
with w_r = finalReward (or +-1 from rollout status).  Prompts that raise the
likelihood of high-reward conversations and lower the likelihood of
negative-reward ones score higher.
"""

from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch

from .. import ops
from ..apo.optimizer import rollout_weight
from ..utils.observability import trace_range
from ..apo.schema import RolloutResult
from ..models.config import ModelConfig, get_config
from ..models.llama import LlamaModel
from .kvcache import PAGE_SIZE, PagedKVCache
from . import tokenizer as tok


def _pad64(n: int) -> int:
    # pad to the v2 attention kernel's 128-row Q-block granularity
    return (n + 127) & ~127


class LlamaBackend:
    def __init__(self, config: ModelConfig | str = "llama-3-8b", device: Optional[str] = None,
                 seed: int = 0, max_seq: int = 2048, micro_batch: int = 8, tp=None, quant: str = "bf16") -> None:
        if isinstance(config, str):
            config = get_config(config)
        if device is None:
            device = "cuda:0" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.config = config
        self.model = LlamaModel(config, device=self.device, seed=seed, tp=tp, quant=quant)
        self.tokenizer = tok.for_vocab(config.vocab_size)
        self.max_seq = min(max_seq, config.max_position)
        self.micro_batch = micro_batch

    # ------------------------------------------------------------------
    # Sequence building
    # ------------------------------------------------------------------
    _ROLE_IDS = {"user": tok.ROLE_USER, "assistant": tok.ROLE_ASSISTANT,
                 "tool": tok.ROLE_TOOL}

    def build_scored_sequence(self, candidate_prompt: str, rollout: RolloutResult
                              ) -> Tuple[List[int], List[bool]]:
        """Token ids + mask marking assistant-content tokens (the scored ones)."""
        ids = [tok.BOS, tok.ROLE_SYSTEM]
        mask = [False, False]
        ids.extend(self.tokenizer.encode(candidate_prompt, max_tokens=600))
        mask.extend([False] * (len(ids) - len(mask)))
        for m in rollout.messages:
            role_id = self._ROLE_IDS.get(m.role, tok.ROLE_USER)
            ids.append(role_id)
            mask.append(False)
            content = self.tokenizer.encode(m.content, max_tokens=400)
            ids.extend(content)
            mask.extend([m.role == "assistant"] * len(content))
            ids.append(tok.EOS)
            mask.append(m.role == "assistant")
        ids = ids[: self.max_seq]
        mask = mask[: self.max_seq]
        return ids, mask

    # ------------------------------------------------------------------
    # Teacher-forced log-prob of masked positions for a batch of sequences
    # ------------------------------------------------------------------
    def sequence_logprobs(self, sequences: Sequence[Tuple[List[int], List[bool]]]
                          ) -> List[float]:
        """Returns the mean per-token logprob of masked tokens per sequence.

        Microbatches are pipelined over two HIP streams (attention/elemwise
        work of one co-schedules with the GEMMs of the next) and synced ONCE
        at the end instead of per-chunk."""
        mb = self.micro_batch
        chunks = [sequences[i: i + mb] for i in range(0, len(sequences), mb)]
        outs = []
        for chunk, stream in zip(chunks, self._stream_cycle(len(chunks))):
            B = len(chunk)
            S = _pad64(max(len(ids) for ids, _ in chunk))
            tokens = torch.zeros(B, S, dtype=torch.long)
            mask = torch.zeros(B, S, dtype=torch.bool)
            for b, (ids, m) in enumerate(chunk):
                tokens[b, : len(ids)] = torch.tensor(ids, dtype=torch.long)
                mask[b, : len(m)] = torch.tensor(m, dtype=torch.bool)
            outs.append(self._on_stream(
                stream, tokens.to(self.device), mask.to(self.device)))
        return torch.cat(outs).cpu().tolist() if outs else []

    def score_microbatches(self, tokens: torch.Tensor, mask: torch.Tensor,
                           micro_batch: Optional[int] = None) -> torch.Tensor:
        """Device-resident scores for [N, S] token/mask tensors, microbatched
        over two streams; ONE host sync when the caller reads the result."""
        mb = micro_batch or self.micro_batch
        N = tokens.shape[0]
        if N == 0:
            return torch.zeros(0, dtype=torch.float32, device=tokens.device)
        idxs = list(range(0, N, mb))
        outs = []
        for i, stream in zip(idxs, self._stream_cycle(len(idxs))):
            outs.append(self._on_stream(stream, tokens[i:i + mb], mask[i:i + mb]))
        return torch.cat(outs)

    # two alternating HIP streams (None entries = current stream / CPU)
    def _stream_cycle(self, n: int):
        if self.device.type != "cuda" or n <= 1:
            return [None] * n
        if getattr(self, "_streams", None) is None:
            self._streams = [torch.cuda.Stream(), torch.cuda.Stream()]
        return [self._streams[i % 2] for i in range(n)]

    def _on_stream(self, stream, tokens, mask):
        if stream is None:
            return self.score_token_batch_device(tokens, mask)
        cur = torch.cuda.current_stream()
        stream.wait_stream(cur)
        with torch.cuda.stream(stream):
            out = self.score_token_batch_device(tokens, mask)
        cur.wait_stream(stream)
        return out

    def score_token_batch(self, tokens: torch.Tensor, mask: torch.Tensor) -> torch.Tensor:
        """As score_token_batch_device, synced to CPU."""
        return self.score_token_batch_device(tokens, mask).cpu()

    def score_token_batch_device(self, tokens: torch.Tensor, mask: torch.Tensor
                                 ) -> torch.Tensor:
        """Mean per-token logprob of masked positions per sequence.

        tokens/mask: [B, S] on the engine device (S % 64 == 0).  Fully
        vectorized: only assistant-position hidden rows go through the
        lm_head GEMM (a ~4x lm_head saving at the bench's 25% mask).
        Returns f32 [B] on the engine device (no host sync).
        """
        B, S = tokens.shape
        # trace_range = rocTX/NVTX span: shows up in rocprofv3 timelines
        # alongside the kernels (SURVEY §5.1's perf-span analog)
        with trace_range(f"score_b{B}_s{S}"):
            # mask processing FIRST: nonzero() host-syncs its stream, so
            # doing it before the prefill is enqueued keeps the two-stream
            # pipeline 2-deep (the sync only waits on work from two
            # microbatches back)
            m = mask.clone()
            m[:, 0] = False  # position 0 has no predictor
            flat_pos = m.reshape(-1).nonzero(as_tuple=False).squeeze(1)
            if flat_pos.numel() == 0:
                return torch.zeros(B, dtype=torch.float32, device=tokens.device)
            hidden = self.model.prefill(tokens)  # [B, S, H]
            # p >= 1 within every sequence (position 0 masked off above), so
            # flat_pos - 1 stays inside the same sequence's rows
            rows = hidden.reshape(B * S, -1)[flat_pos - 1]
            logits = self.model.logits(rows)
            targets = tokens.reshape(-1)[flat_pos].to(torch.int32)
            lp = ops.target_logprob(logits, targets)
            seq_of = torch.div(flat_pos, S, rounding_mode="floor")
            sums = torch.zeros(B, dtype=torch.float32, device=lp.device)
            counts = torch.zeros(B, dtype=torch.float32, device=lp.device)
            sums.index_add_(0, seq_of, lp.float())
            counts.index_add_(0, seq_of, torch.ones_like(lp, dtype=torch.float32))
            return sums / counts.clamp(min=1)

    # ------------------------------------------------------------------
    # PromptOptimizerBackend API
    # ------------------------------------------------------------------
    def score_batch(self, candidate_prompts: Sequence[str],
                    rollouts: Sequence[RolloutResult]) -> List[float]:
        """Score each candidate against all rollouts (one forward per pair)."""
        seqs = []
        for c in candidate_prompts:
            for r in rollouts:
                seqs.append(self.build_scored_sequence(c, r))
        lps = self.sequence_logprobs(seqs)
        weights = [rollout_weight(r) for r in rollouts]
        wnorm = sum(abs(w) for w in weights) or 1.0
        out = []
        R = len(rollouts)
        for ci in range(len(candidate_prompts)):
            s = sum(weights[ri] * lps[ci * R + ri] for ri in range(R))
            out.append(s / wnorm)
        return out

    def score(self, candidate_prompt: str, rollouts: Sequence[RolloutResult]) -> float:
        return self.score_batch([candidate_prompt], rollouts)[0]

    # ------------------------------------------------------------------
    # Decode (persistent cache + hipGraph-captured step on GPU)
    # ------------------------------------------------------------------
    def _decode_state(self):
        """Persistent (cache, graph) reused across generate() calls."""
        if getattr(self, "_decode_cache", None) is None:
            pages = (self.max_seq + PAGE_SIZE - 1) // PAGE_SIZE + 2
            self._decode_cache = PagedKVCache(self.config, pages, self.device,
                                              num_kv_heads=self.model.local_kv_heads)
            self._decode_graph = None
            if self.device.type == "cuda":
                from .graph import DecodeGraph
                self._decode_graph = DecodeGraph(self.model, self._decode_cache, pages)
        return self._decode_cache, self._decode_graph

    def _decode_one(self, cache, graph, seq: int, token: int) -> torch.Tensor:
        if graph is not None:
            return graph.step(token, seq)[0]
        pos = torch.tensor([cache.seq_lens[seq]], device=self.device)
        return self.model.decode_step(torch.tensor([token], device=self.device),
                                      pos, cache, [seq])[0]

    @torch.no_grad()
    def stream_generate(self, prompt: str, max_new_tokens: int,
                        should_stop, on_chunk, *, temperature: float = 0.0,
                        top_p: float = 1.0, sample_seed: Optional[int] = None,
                        stop: Optional[List[str]] = None,
                        presence_penalty: float = 0.0,
                        frequency_penalty: float = 0.0,
                        logit_bias: Optional[dict] = None) -> str:
        """Incremental decode for the transport layer: calls
        on_chunk(cumulative_text) per token, honors should_stop between
        tokens.  temperature <= 0 is greedy (the APO default — edits must
        be deterministic); temperature > 0 samples from the
        temperature-scaled softmax with nucleus (top-p) filtering, seeded
        by ``sample_seed`` for reproducible sampling."""
        # context budget: a decode budget >= max_seq must not go negative
        # (a negative python slice KEPT the whole prompt and overflowed the
        # paged cache — examples/apo_demo.py with max_edit_tokens==max_seq)
        ctx_budget = self.max_seq - max_new_tokens - 8
        if ctx_budget < 32:
            ctx_budget = max(32, self.max_seq // 4)
        ids = [tok.BOS] + self.tokenizer.encode(prompt, max_tokens=ctx_budget)
        ids.append(tok.ROLE_ASSISTANT)
        real = len(ids)
        # decode steps can never push past the cache (+2 slack pages)
        max_new_tokens = max(1, min(max_new_tokens, self.max_seq - real))
        S = _pad64(real)
        cache, graph = self._decode_state()
        # full reset (also recovers pages leaked by an aborted prefill)
        cache._free = list(range(cache.num_pages - 1, -1, -1))
        cache.block_tables, cache.seq_lens = [], []
        seq = cache.new_seq()
        tokens = torch.zeros(1, S, dtype=torch.long)
        tokens[0, :real] = torch.tensor(ids, dtype=torch.long)
        hidden = self.model.prefill(tokens.to(self.device), cache=cache, seqs=[seq],
                                    real_lens=[real])
        last_hidden = hidden[0, real - 1]
        out_ids = []
        text = ""
        gen = None
        if temperature > 0:
            gen = torch.Generator(device=self.device)
            gen.manual_seed(sample_seed if sample_seed is not None else 0)
        for step in range(max_new_tokens):
            if should_stop():
                break
            logits = self.model.logits(last_hidden.reshape(1, -1))
            if logit_bias:
                logits = logits.clone()
                for t, bias in logit_bias.items():
                    logits[0, int(t)] += float(bias)
            if (presence_penalty or frequency_penalty) and out_ids:
                # OpenAI semantics: penalize GENERATED tokens on the logits
                seen = torch.tensor(sorted(set(out_ids)), device=logits.device)
                counts = torch.tensor(
                    [out_ids.count(int(t)) for t in seen],
                    device=logits.device, dtype=logits.dtype)
                logits = logits.clone()
                logits[0, seen] -= (presence_penalty
                                    + frequency_penalty * counts)
            if temperature > 0:
                nxt = _sample_token(logits, temperature, top_p, gen)
            else:
                nxt = int(ops.argmax_rows(logits)[0])
            if nxt == tok.EOS:
                break
            out_ids.append(nxt)
            text = self.tokenizer.decode(out_ids)
            if stop:
                cut = min((text.find(sq) for sq in stop if sq in text),
                          default=-1)
                if cut >= 0:
                    text = text[:cut]
                    on_chunk(text)
                    break
            on_chunk(text)
            last_hidden = self._decode_one(cache, graph, seq, nxt)
        return text

    @torch.no_grad()
    def embed(self, texts: List[str], max_tokens: int = 512) -> List[List[float]]:
        """Sequence embeddings: mean-pooled final hidden states (the
        backbone has no trained embedding head; this is the standard
        pooled-representation fallback, unit-normalized)."""
        out = []
        for text in texts:
            ids = [tok.BOS] + self.tokenizer.encode(text, max_tokens=max_tokens)
            S = _pad64(len(ids))
            t = torch.zeros(1, S, dtype=torch.long)
            t[0, : len(ids)] = torch.tensor(ids, dtype=torch.long)
            hidden = self.model.prefill(t.to(self.device))  # [1, S, H]
            v = hidden[0, : len(ids)].float().mean(dim=0)
            v = v / (v.norm() + 1e-8)
            out.append(v.cpu().tolist())
        return out

    @torch.no_grad()
    def generate(self, prompt: str, max_new_tokens: int = 256, *,
                 temperature: float = 0.0, top_p: float = 1.0,
                 sample_seed: Optional[int] = None) -> str:
        return self.stream_generate(prompt, max_new_tokens,
                                    should_stop=lambda: False,
                                    on_chunk=lambda _t: None,
                                    temperature=temperature, top_p=top_p,
                                    sample_seed=sample_seed)

def _sample_token(logits: torch.Tensor, temperature: float, top_p: float,
                  gen: torch.Generator) -> int:
    """Nucleus sampling from [1, V] logits (f32 softmax for stability)."""
    probs = torch.softmax(logits.float().squeeze(0) / temperature, dim=-1)
    if top_p < 1.0:
        sp, idx = torch.sort(probs, descending=True)
        cum = torch.cumsum(sp, dim=-1)
        keep = cum - sp < top_p  # always keeps the top token
        sp = torch.where(keep, sp, torch.zeros_like(sp))
        sp = sp / sp.sum()
        pick = torch.multinomial(sp, 1, generator=gen)
        return int(idx[pick])
    return int(torch.multinomial(probs, 1, generator=gen))
