"""RCCL (backend "nccl") world-2 tests on a single MI355X.

RCCL permits multiple ranks per device, so the real collective path —
bf16-over-RCCL all_reduce, broadcast_object_list, and the TP=2 forward —
is exercised on ONE leased GPU exactly as it will run across 8.  These are
the hardware burn-in for senweaver_amd/parallel/{dist,tp}.py whose gloo
world-2 CPU twins live in test_dist_cpu.py.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, fn_name, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    torch.cuda.set_device(0)  # both ranks share the single GPU
    dist.init_process_group("nccl", rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        q.put((rank, result))
    finally:
        dist.destroy_process_group()


def _run_dist(fn_name, world=2):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29611 + (os.getpid() % 500)
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name, q))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=600)
    results = {}
    while not q.empty():
        rank, res = q.get()
        results[rank] = res
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    assert len(results) == world
    return results


# --- payload fns (module-level so spawn can pickle by name) ---

def _payload_allreduce_bf16(rank, world):
    dev = torch.device("cuda:0")
    x = torch.full((1024,), float(rank + 1), dtype=torch.bfloat16, device=dev)
    dist.all_reduce(x, op=dist.ReduceOp.SUM)
    torch.cuda.synchronize()
    return x.float().sum().item()


def _payload_scores_and_broadcast(rank, world):
    from senweaver_amd.parallel import broadcast_strings, dp_scores_allreduce
    dev = torch.device("cuda:0")
    my_idx = list(range(rank, 16, world))
    my_scores = [float(100 * rank + i) for i in range(len(my_idx))]
    scores = dp_scores_allreduce(16, my_idx, my_scores, dev)
    strings = ["cand-a", "cand-b"] if rank == 0 else None
    got = broadcast_strings(strings, src=0)
    return scores, got


def _payload_tp2_forward(rank, world):
    from senweaver_amd.models import tiny_tp
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.tp import TPContext

    tp = TPContext.from_default_group()
    model = LlamaModel(tiny_tp(), device="cuda:0", seed=5, tp=tp)
    tokens = torch.randint(0, 512, (1, 64),
                           generator=torch.Generator().manual_seed(9)).cuda()
    hidden = model.prefill(tokens)
    torch.cuda.synchronize()
    return hidden.float().sum(-1).squeeze(0).cpu().tolist()[:8]


def test_rccl_allreduce_world2_one_gpu():
    results = _run_dist("_payload_allreduce_bf16")
    # sum over ranks: (1+2) * 1024
    for rank, total in results.items():
        assert total == pytest.approx(3.0 * 1024, rel=1e-3)


def test_rccl_beam_collectives_world2():
    results = _run_dist("_payload_scores_and_broadcast")
    s0, b0 = results[0]
    s1, b1 = results[1]
    assert s0 == pytest.approx(s1, abs=0)
    assert b0 == b1 == ["cand-a", "cand-b"]


def test_tp2_over_rccl_matches_tp1():
    """Config-5 burn-in: the TP sharded forward over real RCCL on device,
    asserted against the unsharded single-GPU model."""
    from senweaver_amd.models import tiny_tp
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_tp(), device="cuda:0", seed=5)
    tokens = torch.randint(0, 512, (1, 64),
                           generator=torch.Generator().manual_seed(9)).cuda()
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).cpu().tolist()[:8]
    del ref_model
    torch.cuda.empty_cache()

    results = _run_dist("_payload_tp2_forward")
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
    assert results[0] == pytest.approx(results[1], abs=1e-5)
