"""Weight checkpointing: save/load the backbone as a safetensors file.

The engine normally random-initializes from a shared seed (no network for
checkpoints in this environment), but a real framework needs durable
weights: trained/converted checkpoints, cross-process distribution beyond
seed agreement, and restart without re-init.  The file stores EXACTLY the
tensors the model holds — bf16 projections, or the fp8/mxfp8 `_q`/`_s`
pairs for quantized models (whose bf16 originals are deleted at build) —
plus metadata (model name, quant mode, TP/EP shard coordinates) so a
mismatched restore fails loudly instead of silently misloading.
"""

from __future__ import annotations

import torch


def _model_tensors(model) -> dict:
    t = {"embed": model.embed, "final_norm": model.final_norm}
    if model.lm_head is not model.embed:
        t["lm_head"] = model.lm_head
    for name in ("lm_head_q", "lm_head_s"):
        if hasattr(model, name):
            t[name] = getattr(model, name)
    for i, L in enumerate(model.layers):
        for k, v in L.items():
            t[f"layers.{i}.{k}"] = v
    return t


def save_weights(model, path: str) -> int:
    """Write the model's tensors to ``path`` (safetensors).  Returns the
    tensor count.  TP/EP-sharded models save THEIR shard; metadata records
    the shard coordinates."""
    from safetensors.torch import save_file

    tensors = {k: v.detach().cpu().contiguous()
               for k, v in _model_tensors(model).items()}
    meta = {
        "model": model.config.name,
        "quant": model.quant,
        "tp_world": str(model.tp.world), "tp_rank": str(model.tp.rank),
        "ep_world": str(model.ep.world), "ep_rank": str(model.ep.rank),
        "format": "senweaver_amd.v1",
    }
    save_file(tensors, path, metadata=meta)
    return len(tensors)


def load_weights(model, path: str) -> int:
    """Restore tensors saved by save_weights into ``model`` (same config,
    quant mode, and shard coordinates).  Returns the tensor count.

    Call BEFORE capturing a decode graph: restoration copies in place so
    live captures stay valid, but a shape mismatch raises first.
    """
    from safetensors import safe_open

    targets = _model_tensors(model)
    n = 0
    with safe_open(path, framework="pt") as f:
        meta = f.metadata() or {}
        for key, want in (("model", model.config.name), ("quant", model.quant),
                          ("tp_world", str(model.tp.world)),
                          ("tp_rank", str(model.tp.rank)),
                          ("ep_world", str(model.ep.world)),
                          ("ep_rank", str(model.ep.rank))):
            got = meta.get(key)
            if got is not None and got != want:
                raise ValueError(
                    f"checkpoint {key}={got!r} does not match model {want!r}")
        keys = set(f.keys())
        missing = set(targets) - keys
        extra = keys - set(targets)
        if missing or extra:
            raise ValueError(f"checkpoint/model tensor mismatch: "
                             f"missing={sorted(missing)[:4]} extra={sorted(extra)[:4]}")
        for key in f.keys():
            src = f.get_tensor(key)
            dst = targets[key]
            if tuple(src.shape) != tuple(dst.shape) or src.dtype != dst.dtype:
                raise ValueError(f"tensor {key}: checkpoint {tuple(src.shape)}/"
                                 f"{src.dtype} vs model {tuple(dst.shape)}/{dst.dtype}")
            with torch.no_grad():
                dst.copy_(src.to(dst.device))
            n += 1
    return n
