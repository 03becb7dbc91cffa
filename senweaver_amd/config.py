"""Unified engine configuration: one JSON/YAML file + environment overrides.

The reference scatters tuning constants across per-service config objects
persisted in storage (APOConfig apoService.ts:279-292, upload config
traceCollectorService.ts:315-329), product.json, and compile-time constant
modules (tokenOptimizationConfig.ts, SMART_CONTEXT_CONFIG).  Per SURVEY §5.6
the rebuild centralizes them: ``EngineConfig.load`` reads an optional config
file, then applies ``SENWEAVER_*`` environment overrides, reproducing the
reference's default values verbatim.

Env override format: SENWEAVER_<SECTION>_<KEY>=value, e.g.
SENWEAVER_APO_BEAMWIDTH=8, SENWEAVER_TRACE_MAXTRACES=500.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional

from .apo.schema import default_apo_config


def default_trace_config() -> Dict[str, Any]:
    return {
        "maxContentPreview": 500,
        "maxTraces": 1000,
        "maxSpansPerTrace": 200,
        "flushIntervalMs": 30000,
        "autoUploadIntervalMs": 300000,
    }


def default_context_config() -> Dict[str, Any]:
    return {
        "charsPerToken": 3.5,
        "trimToLen": 500,
        "keepRecentCount": 10,
        "overflowThreshold": 0.55,
        "pruneProtectTokens": 20000,
        "largeOutputThreshold": 50000,
        "systemMsgMaxRatio": 0.30,
        "systemMsgHardCap": 60000,
        "apoRulesMaxChars": 2000,
    }


def default_engine_config() -> Dict[str, Any]:
    return {
        "model": "llama-3-8b",
        "maxSeq": 2048,
        "microBatch": 8,
        "quant": "bf16",
        "tensorParallel": 1,
    }


@dataclass
class EngineConfig:
    apo: Dict[str, Any] = field(default_factory=default_apo_config)
    trace: Dict[str, Any] = field(default_factory=default_trace_config)
    context: Dict[str, Any] = field(default_factory=default_context_config)
    engine: Dict[str, Any] = field(default_factory=default_engine_config)

    @classmethod
    def load(cls, path: Optional[str] = None, env: Optional[Dict[str, str]] = None) -> "EngineConfig":
        cfg = cls()
        if path and os.path.exists(path):
            try:
                if path.endswith((".yaml", ".yml")):
                    import yaml
                    data = yaml.safe_load(open(path)) or {}
                else:
                    data = json.load(open(path))
                for section in ("apo", "trace", "context", "engine"):
                    if isinstance(data.get(section), dict):
                        getattr(cfg, section).update(data[section])
            except (OSError, ValueError) as e:
                raise ValueError(f"bad config file {path}: {e}") from e
        cfg._apply_env(env if env is not None else dict(os.environ))
        return cfg

    def _apply_env(self, env: Dict[str, str]) -> None:
        for key, value in env.items():
            if not key.startswith("SENWEAVER_"):
                continue
            parts = key[len("SENWEAVER_"):].split("_", 1)
            if len(parts) != 2:
                continue
            section = parts[0].lower()
            if section not in ("apo", "trace", "context", "engine"):
                continue
            target = getattr(self, section)
            # case-insensitive key match against the section's known keys
            match = next((k for k in target if k.lower() == parts[1].lower()), None)
            if match is None:
                continue
            old = target[match]
            if isinstance(old, bool):
                target[match] = value.strip().lower() in ("1", "true", "yes")
            elif isinstance(old, int):
                target[match] = int(value)
            elif isinstance(old, float):
                target[match] = float(value)
            else:
                target[match] = value

    def to_json(self) -> Dict[str, Any]:
        return {"apo": self.apo, "trace": self.trace, "context": self.context,
                "engine": self.engine}
