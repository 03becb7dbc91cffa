"""Multi-process (gloo, world_size=2) tests for the candidate-parallel path.

These run on CPU here; the identical code path runs over RCCL ("nccl"
backend) on the 8-GPU xGMI node.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _worker(rank, world, port, fn_name, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        q.put((rank, result))
    finally:
        dist.destroy_process_group()


def _run_dist(fn_name, world=2):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29511 + (os.getpid() % 500)
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    results = {}
    while not q.empty():
        rank, res = q.get()
        results[rank] = res
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    assert len(results) == world
    return results


# --- payload fns (module-level so spawn can pickle by name) ---

def _payload_allreduce_scores(rank, world):
    from senweaver_amd.parallel import dp_scores_allreduce
    my_idx = list(range(rank, 5, world))
    my_scores = [float(10 * i + 1) for i in my_idx]
    return dp_scores_allreduce(5, my_idx, my_scores, torch.device("cpu"))


def _payload_broadcast(rank, world):
    from senweaver_amd.parallel import broadcast_strings
    strings = ["alpha", "beta", "gamma"] if rank == 0 else None
    return broadcast_strings(strings, src=0)


def _payload_candidate_parallel(rank, world):
    from senweaver_amd.apo.optimizer import StubBackend
    from senweaver_amd.apo.schema import RolloutMessage, RolloutResult
    from senweaver_amd.parallel import CandidateParallelScorer

    class CPUStub(StubBackend):
        device = torch.device("cpu")

        def score_batch(self, prompts, rollouts):
            return [self.score(p, rollouts) for p in prompts]

    r = RolloutResult(
        trace_id="t", thread_id="th", status="succeeded", final_reward=1.0,
        reward_dimensions=[], chat_mode="normal",
        messages=[RolloutMessage("assistant", "x")],
        tool_call_stats={"totalCalls": 0, "succeeded": 0, "failed": 0,
                         "successRate": None, "byToolName": {}, "totalDurationMs": 0},
        llm_stats={"totalCalls": 1, "totalTokens": 1},
    )
    backend = CPUStub(seed=7)
    scorer = CandidateParallelScorer(backend)
    prompts = [f"- rule {i}" for i in range(6)]
    scores = scorer(prompts, [r])
    serial = [StubBackend(seed=7).score(p, [r]) for p in prompts]
    return (scores, serial)


def test_dp_scores_allreduce_world2():
    results = _run_dist("_payload_allreduce_scores")
    for rank, vec in results.items():
        assert vec == [1.0, 11.0, 21.0, 31.0, 41.0]


def test_broadcast_strings_world2():
    results = _run_dist("_payload_broadcast")
    assert results[0] == results[1] == ["alpha", "beta", "gamma"]


def test_candidate_parallel_scorer_world2():
    results = _run_dist("_payload_candidate_parallel")
    for rank, (scores, serial) in results.items():
        assert scores == pytest.approx(serial, abs=1e-6)  # f32 collective rounding
    # both ranks computed the identical full vector
    assert results[0][0] == pytest.approx(results[1][0], abs=0)


def _payload_tp_prefill(rank, world):
    import torch
    from senweaver_amd.models import tiny_tp
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.tp import TPContext

    tp = TPContext.from_default_group()
    model = LlamaModel(tiny_tp(), device="cpu", seed=5, tp=tp)
    tokens = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(9))
    hidden = model.prefill(tokens)
    return hidden.float().sum(-1).squeeze(0).tolist()[:8]


def test_tp2_matches_tp1():
    """TP=2 over gloo reproduces the single-rank model (same seed/full weights)."""
    import torch
    from senweaver_amd.models import tiny_tp
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_tp(), device="cpu", seed=5)
    tokens = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(9))
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).tolist()[:8]

    results = _run_dist("_payload_tp_prefill")
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
    assert results[0] == pytest.approx(results[1], abs=1e-5)


def _payload_distributed_beam(rank, world):
    """Full distributed beam round: rank-0 expansion broadcast + sharded
    scoring + all-reduced Top-K -> identical beam state on every rank."""
    from senweaver_amd.apo import APOService, BeamSearchEngine
    from senweaver_amd.apo.optimizer import StubBackend
    from senweaver_amd.parallel import CandidateParallelScorer, broadcast_strings
    from senweaver_amd.storage import MemoryStorage
    from senweaver_amd.trace import TraceCollector
    import torch

    class CPUStub(StubBackend):
        device = torch.device("cpu")

        def score_batch(self, prompts, rollouts):
            return [self.score(p, rollouts) for p in prompts]

    backend = CPUStub(seed=11)
    storage = MemoryStorage()
    tc = TraceCollector(storage=storage, clock=lambda: 1700000000000,
                        uuid_fn=lambda: f"u{id(object()) % 97}")
    n = {"i": 0}

    def uuid_fn():
        n["i"] += 1
        return f"uuid-{n['i']}"

    tc._uuid = uuid_fn
    for i in range(4):
        th = f"th{i}"
        tid = tc.start_trace(th, {"chatMode": "agent"})
        tc.record_user_message(th, 0, f"task {i}")
        tc.record_assistant_message(th, 1, f"done {i}")
        tc.end_trace(tid)
        tc.record_user_feedback(th, 1, "good" if i % 2 else "bad")
    apo = APOService(tc, storage=storage, clock=lambda: 1700000000000, uuid_fn=uuid_fn)

    def expand_fn(parent_contents, rollouts, branch_factor):
        if rank == 0:
            out = []
            for p in parent_contents:
                for b in range(branch_factor):
                    out.append(backend.generate(f"{p}::variant{b}"))
        else:
            out = None
        return broadcast_strings(out, src=0)

    engine = BeamSearchEngine(backend, score_fn=CandidateParallelScorer(backend),
                              expand_fn=expand_fn)
    engine.run_round(apo)
    engine.run_round(apo)
    state = apo.get_beam_state()
    return {
        "round": state.current_round,
        "beam": [(b.version, b.content[:40], round(b.score, 6)) for b in state.beam],
        "best": state.history_best_prompt.version,
        "best_score": round(state.history_best_score, 6),
    }


def test_distributed_beam_identical_state():
    results = _run_dist("_payload_distributed_beam")
    assert results[0] == results[1]
    assert results[0]["round"] == 2
    assert len(results[0]["beam"]) == 4


def _payload_tp_prefill_fp8(rank, world):
    import torch
    from senweaver_amd.models import tiny_tp
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.tp import TPContext

    tp = TPContext.from_default_group()
    model = LlamaModel(tiny_tp(), device="cpu", seed=5, tp=tp, quant="fp8")
    ref16 = LlamaModel(tiny_tp(), device="cpu", seed=5)
    tokens = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(9))
    hidden = model.prefill(tokens).float()
    h16 = ref16.prefill(tokens).float()
    rel = ((hidden - h16).norm() / h16.norm()).item()
    return [rel] + hidden.sum(-1).squeeze(0).tolist()[:8]


def test_tp2_fp8_agrees_and_tracks_bf16():
    """config-5 cross: TP sharding composed with fp8 projections.

    Row-parallel shards (o/down) quantize over their LOCAL K-half, so fp8
    TP=k is a different (finer) quantization grouping than TP=1 — bitwise
    equality is not expected.  The contract: every rank produces the
    identical result (collective determinism) and the fp8 TP output stays
    within fp8 tolerance of the bf16 model."""
    results = _run_dist("_payload_tp_prefill_fp8")
    for rank, vals in results.items():
        assert vals[0] < 0.15  # rel vs bf16
    assert results[0] == pytest.approx(results[1], abs=1e-5)


def _payload_moe_tp_prefill(rank, world):
    import torch
    from senweaver_amd.models import tiny_moe_tp
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.tp import TPContext

    tp = TPContext.from_default_group()
    model = LlamaModel(tiny_moe_tp(), device="cpu", seed=6, tp=tp)
    tokens = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(11))
    hidden = model.prefill(tokens)
    return hidden.float().sum(-1).squeeze(0).tolist()[:8]


def test_moe_tp2_matches_tp1():
    """MoE + TP: per-expert Megatron sharding reproduces the TP=1 model
    (replicated router -> identical routing; column gate|up + row down)."""
    import torch
    from senweaver_amd.models import tiny_moe_tp
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_moe_tp(), device="cpu", seed=6)
    tokens = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(11))
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).tolist()[:8]

    results = _run_dist("_payload_moe_tp_prefill")
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
    assert results[0] == pytest.approx(results[1], abs=1e-5)


def test_bench_torchrun_world2_contract():
    """Launch bench.py through torch.distributed.run exactly as the driver
    does (world 2, gloo on CPU, tiny model): rank 0 must print one JSON line
    with the contract fields and dp2 parallelism."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29641", "bench.py", "--gpus", "2", "--steps", "1",
         "--warmup", "0", "--model", "tiny-debug", "--seq-len", "256"],
        cwd=repo, capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["metric"].startswith("APO iterations/sec")
    assert d["n_gpus"] == 2 and d["scaling"] == "strong"
    assert d["config"]["parallelism"] == "dp2"
    assert d["value"] > 0 and d["higher_is_better"] is True


def _payload_ep_moe(rank, world):
    import torch
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.ep import EPContext

    ep = EPContext.from_default_group()
    model = LlamaModel(tiny_moe(), device="cpu", seed=11, ep=ep)
    tokens = torch.randint(0, 512, (1, 48),
                           generator=torch.Generator().manual_seed(4))
    hidden = model.prefill(tokens)
    return hidden.float().sum(-1).squeeze(0).tolist()[:8]


def test_ep2_matches_replicated():
    """EP=2 (expert-sharded, all-to-all routed) reproduces the replicated
    single-rank MoE forward (same seed => same full weights)."""
    import torch
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_moe(), device="cpu", seed=11)
    tokens = torch.randint(0, 512, (1, 48),
                           generator=torch.Generator().manual_seed(4))
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).tolist()[:8]

    results = _run_dist("_payload_ep_moe")
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
    assert results[0] == pytest.approx(results[1], abs=1e-5)


def _payload_tp2_ep2_grid(rank, world):
    import torch
    from senweaver_amd.models import tiny_moe_tp
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel import build_tp_ep_grid

    tp, ep = build_tp_ep_grid(2, 2)
    model = LlamaModel(tiny_moe_tp(), device="cpu", seed=13, tp=tp, ep=ep)
    tokens = torch.randint(0, 512, (1, 48),
                           generator=torch.Generator().manual_seed(6))
    hidden = model.prefill(tokens)
    return hidden.float().sum(-1).squeeze(0).tolist()[:8]


def test_tp2_ep2_grid_matches_unsharded():
    """World 4 as a TP2 x EP2 grid (subgroups): attention sharded over the
    TP pair, experts sharded over the EP pair, result identical to the
    unsharded single-rank forward."""
    import torch
    from senweaver_amd.models import tiny_moe_tp
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_moe_tp(), device="cpu", seed=13)
    tokens = torch.randint(0, 512, (1, 48),
                           generator=torch.Generator().manual_seed(6))
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).tolist()[:8]

    results = _run_dist("_payload_tp2_ep2_grid", world=4)
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
    assert results[0] == pytest.approx(results[3], abs=1e-5)


def _payload_ep_dispatch_empty_shard(rank, world):
    import torch
    from senweaver_amd.parallel.ep import EPContext

    ep = EPContext.from_default_group()
    E = 4  # rank0 owns experts 0-1, rank1 owns 2-3
    # every token on every rank routes to experts 0/1 -> rank1's shard is EMPTY
    counts = torch.tensor([4, 2, 0, 0])
    x = (torch.arange(6 * 3, dtype=torch.float32).reshape(6, 3)
         + 1000 * rank)
    x_local, local_counts, meta = ep.dispatch(x, counts, E)
    if rank == 0:
        assert x_local.shape[0] == 12  # both ranks' 6 rows
        assert local_counts.tolist() == [8, 4]
    else:
        assert x_local.shape[0] == 0
        assert local_counts.tolist() == [0, 0]
    y = ep.combine(x_local * 2, meta)
    # roundtrip: every rank gets back exactly its own rows, doubled
    return torch.allclose(y, x * 2), y.shape[0]


def test_ep_dispatch_empty_shard():
    """EP all-to-all with a rank that owns zero routed tokens: dispatch,
    the empty-side grouped compute, and combine must all roundtrip."""
    results = _run_dist("_payload_ep_dispatch_empty_shard")
    for rank, (ok, rows) in results.items():
        assert ok and rows == 6
