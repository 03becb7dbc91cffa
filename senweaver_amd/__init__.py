"""senweaver_amd — MI355X-native online-RL / Automatic Prompt Optimization engine.

A brand-new framework with the capabilities of SenWeaver-IDE's RL pipeline
(TraceCollectorService -> 9-dimension reward -> APOService), rebuilt MI355X-first:

- ``senweaver_amd.trace``    — conversation-turn trace spans + 9-dim weighted reward
  (format-compatible with reference ``common/traceCollectorService.ts``).
- ``senweaver_amd.apo``      — effectiveness reports, 6-pattern failure detector,
  suggestion lifecycle, textual-gradient + beam-search prompt optimization
  (format-compatible with reference ``common/apoService.ts``).
- ``senweaver_amd.ops``      — hand-written CDNA4 HIP kernels (MFMA GEMM, RMSNorm,
  RoPE, paged/flash attention, sampling) for gfx950.
- ``senweaver_amd.models``   — Llama / Mixtral backbones running on those kernels.
- ``senweaver_amd.engine``   — KV cache, generation, log-prob scoring.
- ``senweaver_amd.parallel`` — RCCL-over-xGMI candidate-parallel beam scoring,
  tensor parallelism, expert parallelism and the TP x EP subgroup grid.
- ``senweaver_amd.server``   — native C++ daemon, engine worker, launcher CLI,
  and the OpenAI-compatible HTTP serving surface.

The reference's "model" is a remote LLM over HTTPS; here the critique generator and
beam scorer run locally on MI355X GPUs.
"""

__version__ = "0.2.0"
