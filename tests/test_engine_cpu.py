"""End-to-end engine tests on CPU with the tiny debug model.

These exercise the same code path the MI355X runs (ops dispatch to the fp32
torch references on CPU) so the model/cache/scorer logic is verified here
and only kernel numerics remain for the GPU suite.
"""

import pytest
import torch

from senweaver_amd.apo import APOService, BeamSearchEngine, LocalGradientEngine
from senweaver_amd.engine import LlamaBackend, PagedKVCache
from senweaver_amd.engine import tokenizer as tok
from senweaver_amd.models import tiny_debug
from senweaver_amd.models.llama import LlamaModel
from senweaver_amd.storage import MemoryStorage
from senweaver_amd.trace import TraceCollector


@pytest.fixture(scope="module")
def backend():
    return LlamaBackend(tiny_debug(), device="cpu", max_seq=256, micro_batch=4)


def test_prefill_shapes(backend):
    tokens = torch.randint(0, 512, (2, 64))
    h = backend.model.prefill(tokens)
    assert h.shape == (2, 64, 256)
    assert h.dtype == torch.bfloat16
    assert torch.isfinite(h.float()).all()


def test_prefill_decode_consistency():
    """Greedy continuation via paged decode must match re-running prefill."""
    cfg = tiny_debug()
    backend = LlamaBackend(cfg, device="cpu", max_seq=256)
    model = backend.model
    prompt = torch.randint(5, 500, (1, 37))

    # path A: prefill(prompt) -> decode 3 tokens through the paged cache
    cache = PagedKVCache(cfg, num_pages=16, device="cpu")
    seq = cache.new_seq()
    import torch.nn.functional as F
    S = 64
    padded = torch.zeros(1, S, dtype=torch.long)
    padded[0, :37] = prompt
    h = model.prefill(padded, cache=cache, seqs=[seq], real_lens=[37])
    last = h[0, 36]
    toks_a = []
    for step in range(3):
        logits = model.logits(last.reshape(1, -1))
        nxt = int(logits.float().argmax())
        toks_a.append(nxt)
        last = model.decode_step(torch.tensor([nxt]), torch.tensor([37 + step]),
                                 cache, [seq])[0]

    # path B: full prefill over prompt+generated each step (no cache)
    cur = prompt.clone()
    toks_b = []
    for _ in range(3):
        L = cur.shape[1]
        Sp = (L + 63) & ~63
        pad = torch.zeros(1, Sp, dtype=torch.long)
        pad[0, :L] = cur
        h = model.prefill(pad)
        logits = model.logits(h[0, L - 1].reshape(1, -1))
        nxt = int(logits.float().argmax())
        toks_b.append(nxt)
        cur = torch.cat([cur, torch.tensor([[nxt]])], dim=1)

    assert toks_a == toks_b


def test_generate_deterministic(backend):
    out1 = backend.generate("improve the prompt rules", max_new_tokens=8)
    out2 = backend.generate("improve the prompt rules", max_new_tokens=8)
    assert out1 == out2


def test_score_batch_shape_and_determinism(backend):
    from senweaver_amd.apo.schema import RolloutMessage, RolloutResult
    rollouts = []
    for i, (status, reward) in enumerate([("succeeded", 0.8), ("failed", -0.6)]):
        rollouts.append(RolloutResult(
            trace_id=f"t{i}", thread_id=f"th{i}", status=status, final_reward=reward,
            reward_dimensions=[], chat_mode="normal",
            messages=[RolloutMessage("user", "please fix the build"),
                      RolloutMessage("assistant", "I fixed the build by editing main")],
            tool_call_stats={"totalCalls": 0, "succeeded": 0, "failed": 0,
                             "successRate": None, "byToolName": {}, "totalDurationMs": 0},
            llm_stats={"totalCalls": 1, "totalTokens": 100},
        ))
    s1 = backend.score_batch(["- be concise", "- verify everything twice"], rollouts)
    s2 = backend.score_batch(["- be concise", "- verify everything twice"], rollouts)
    assert len(s1) == 2
    assert s1 == s2
    assert all(isinstance(x, float) for x in s1)


def test_full_apo_pipeline_with_llama_backend(fixed_clock, seq_uuid, backend):
    """The reference's end-to-end loop with the local model as optimizer:
    traces -> report -> textual gradient -> beam search -> rule injection."""
    storage = MemoryStorage()
    tc = TraceCollector(storage=storage, clock=fixed_clock, uuid_fn=seq_uuid)
    grad_engine = LocalGradientEngine(backend, max_critique_tokens=8, max_edit_tokens=16)
    apo = APOService(tc, storage=storage, clock=fixed_clock, uuid_fn=seq_uuid,
                     optimizer=grad_engine)
    # seed traces
    for i in range(4):
        th = f"th{i}"
        tid = tc.start_trace(th, {"chatMode": "normal"})
        tc.record_user_message(th, 0, "do the thing")
        tc.record_assistant_message(th, 1, "done the thing")
        tc.end_trace(tid)
        tc.record_user_feedback(th, 1, "good" if i % 2 else "bad")
    tg = apo.request_textual_gradient()
    assert tg is not None and tg.critique

    engine = BeamSearchEngine(backend, max_critique_tokens=8, max_edit_tokens=16)
    state = engine.run_round(apo)
    assert state["round"] == 1
    beam = apo.get_beam_state()
    assert beam is not None and len(beam.beam) >= 1
    assert beam.history_best_prompt is not None
    # scores came from the model (floats, finite)
    assert all(b.score is not None for b in beam.beam)

    from senweaver_amd.apo import inject_rules
    msg = inject_rules("SYS", apo.get_optimized_rules())
    assert len(msg) <= len("SYS") + 2000 + 100


def test_candidate_parallel_scorer_single_rank(backend):
    from senweaver_amd.apo.schema import RolloutMessage, RolloutResult
    from senweaver_amd.parallel import CandidateParallelScorer
    r = RolloutResult(
        trace_id="t", thread_id="th", status="succeeded", final_reward=0.5,
        reward_dimensions=[], chat_mode="normal",
        messages=[RolloutMessage("assistant", "answer text here")],
        tool_call_stats={"totalCalls": 0, "succeeded": 0, "failed": 0,
                         "successRate": None, "byToolName": {}, "totalDurationMs": 0},
        llm_stats={"totalCalls": 1, "totalTokens": 10},
    )
    scorer = CandidateParallelScorer(backend)
    scores = scorer(["- a", "- b", "- c"], [r])
    direct = backend.score_batch(["- a", "- b", "- c"], [r])
    assert scores == pytest.approx(direct, abs=1e-6)


def test_kvcache_page_reuse_and_exhaustion():
    """Pages freed by free_seq are reused; exhaustion raises loudly (the
    'out of pages' path that once leaked pages from aborted prefills)."""
    from senweaver_amd.models import tiny_debug
    cfg = tiny_debug()
    cache = PagedKVCache(cfg, num_pages=4, device="cpu")
    import torch as t

    def fill(seq, n):
        k = t.zeros(n, cfg.num_kv_heads, cfg.head_dim, dtype=t.bfloat16)
        for layer in range(cfg.num_layers):
            cache.append(layer, seq, k, k.clone(),
                         advance_len=(layer == cfg.num_layers - 1))

    s1 = cache.new_seq()
    fill(s1, 16 * 4)  # consumes all 4 pages
    with pytest.raises(RuntimeError):
        s2 = cache.new_seq()
        fill(s2, 16)
    cache.free_seq(s1)
    s3 = cache.new_seq()
    fill(s3, 16 * 2)  # reuses freed pages
    assert cache.ctx_lens_tensor([s3]).tolist() == [32]


def test_swiglu_gemv_cpu_fallback():
    """ops.swiglu_gemv on CPU tensors runs the two-op reference path."""
    import torch
    from senweaver_amd import ops
    gu = torch.randn(1, 2 * 8192, dtype=torch.bfloat16)
    w = torch.randn(64, 8192, dtype=torch.bfloat16)
    got = ops.swiglu_gemv(gu, w)
    g = gu[:, :8192].float()
    u = gu[:, 8192:].float()
    ref = (g * torch.sigmoid(g) * u) @ w.float().t()
    torch.testing.assert_close(got.float(), ref, atol=2.0, rtol=2e-2)


def test_grid_world1():
    """build_tp_ep_grid degrades to identity contexts without dist init."""
    from senweaver_amd.parallel import build_tp_ep_grid
    tp, ep = build_tp_ep_grid(1, 1)
    assert tp.world == 1 and ep.world == 1
    assert ep.local_experts(8) == (0, 8)


def test_generate_budget_clamp():
    """A decode budget >= max_seq must not overflow the paged cache (the
    negative context-budget slice kept the full prompt before)."""
    from senweaver_amd.engine.scorer import LlamaBackend
    from senweaver_amd.models import tiny_debug
    b = LlamaBackend(tiny_debug(), device="cpu", max_seq=128)
    out = b.generate("long prompt " * 200, max_new_tokens=128)
    assert isinstance(out, str)


def test_gemv_quantized_cpu_fallbacks():
    """gemv_fp8w / gemv_mxfp8w off-GPU run the dequant reference path."""
    import torch
    from senweaver_amd import ops
    x = torch.randn(1, 2048, dtype=torch.bfloat16)
    w = torch.randn(64, 2048, dtype=torch.bfloat16)
    wq, ws = ops.quant_fp8(w)
    got = ops.gemv_fp8w(x, wq, ws)
    wf = wq.view(torch.float8_e4m3fn).float() * ws.unsqueeze(1)
    torch.testing.assert_close(got.float(), x.float() @ wf.t(),
                               atol=1.0, rtol=2e-2)
    mq, ms = ops.quant_mxfp8(w)
    got2 = ops.gemv_mxfp8w(x, mq, ms)
    f = mq.view(torch.float8_e4m3fn).float().view(64, 2048 // 32, 32)
    wf2 = (f * torch.exp2(ms.float() - 127).unsqueeze(-1)).reshape(64, 2048)
    torch.testing.assert_close(got2.float(), x.float() @ wf2.t(),
                               atol=1.0, rtol=2e-2)


def test_sampling_decode():
    """temperature=0 stays greedy; temperature>0 is seed-deterministic and
    top-p nucleus filtering is honored."""
    from senweaver_amd.engine.scorer import LlamaBackend, _sample_token
    import torch
    b = LlamaBackend("tiny-debug", device="cpu", max_seq=128)
    g0 = b.generate("sample me", max_new_tokens=6)
    assert g0 == b.generate("sample me", max_new_tokens=6)  # greedy determinism
    s1 = b.generate("sample me", max_new_tokens=6, temperature=0.8, sample_seed=1)
    s1b = b.generate("sample me", max_new_tokens=6, temperature=0.8, sample_seed=1)
    assert s1 == s1b  # seeded sampling determinism
    outs = {b.generate("sample me", max_new_tokens=6, temperature=1.2,
                       sample_seed=sd) for sd in range(6)}
    assert len(outs) > 1  # different seeds explore

    # top-p -> 0 collapses to argmax
    logits = torch.tensor([[0.1, 3.0, 0.2, 2.9]])
    gen = torch.Generator().manual_seed(0)
    assert _sample_token(logits, 1.0, 1e-6, gen) == 1
    # nucleus excludes tail tokens entirely
    gen2 = torch.Generator().manual_seed(0)
    picks = {_sample_token(torch.tensor([[10.0, 9.9, -5.0, -5.0]]), 1.0, 0.9,
                           torch.Generator().manual_seed(sd)) for sd in range(20)}
    assert picks <= {0, 1}
    del gen2


def test_stop_sequences():
    """Generation halts at the first stop string and truncates it away."""
    from senweaver_amd.engine.scorer import LlamaBackend
    b = LlamaBackend("tiny-debug", device="cpu", max_seq=128)
    full = b.generate("halt test", max_new_tokens=8)
    if len(full) < 4:
        return  # degenerate tiny-model output; nothing to split on
    stop_tok = full[2:4]
    out = b.stream_generate("halt test", 8, lambda: False, lambda t: None,
                            stop=[stop_tok])
    assert stop_tok not in out
    assert full.startswith(out)


def test_repetition_penalties_break_loops():
    """Greedy decode of the tiny random-init model loops on a few tokens;
    a frequency penalty must break the loop (OpenAI logit semantics)."""
    from senweaver_amd.engine.scorer import LlamaBackend
    b = LlamaBackend("tiny-debug", device="cpu", max_seq=128)
    ids_plain, ids_pen = [], []
    b.stream_generate("loop breaker", 10, lambda: False,
                      lambda t: None)  # warm path
    # capture token streams via the tokenizer-decoded text uniqueness
    plain = b.generate("loop breaker", max_new_tokens=10)
    pen = b.stream_generate("loop breaker", 10, lambda: False, lambda t: None,
                            frequency_penalty=5.0)
    uniq = lambda s: len(set(s.split()))
    assert uniq(pen) > uniq(plain) or plain != pen


def test_logit_bias_forces_token():
    """A +1000 bias on one token makes greedy pick it every step."""
    from senweaver_amd.engine.scorer import LlamaBackend
    b = LlamaBackend("tiny-debug", device="cpu", max_seq=128)
    out = b.stream_generate("bias me", 5, lambda: False, lambda t: None,
                            logit_bias={300: 1000.0})
    assert out.split() == ["tok300"] * 5
