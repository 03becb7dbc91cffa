"""Settings service — provider/model/chatMode state with persistence.

Capability-compatible with the reference's SenweaverSettingsService
(common/senweaverSettingsService.ts + senweaverSettingsTypes.ts): chat modes
('normal' | 'agent' | 'designer' | 'gather'), per-feature model selection
(Chat / Autocomplete / Apply / SCM), global settings (auto-approve classes,
etc.), JSON persistence.  Providers here are local backbones instead of the
reference's 20 remote HTTP providers.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from ..storage import MemoryStorage
from ..utils.jsonutil import js_parse, js_stringify

CHAT_MODES = ("normal", "agent", "designer", "gather")
FEATURES = ("Chat", "Autocomplete", "Apply", "SCM")
SETTINGS_KEY = "senweaver.settings.state"


@dataclass
class ModelSelection:
    provider_name: str
    model_name: str

    def to_json(self):
        return {"providerName": self.provider_name, "modelName": self.model_name}


@dataclass
class GlobalSettings:
    auto_approve: Dict[str, bool] = field(default_factory=dict)
    chat_mode: str = "agent"
    enable_autocomplete: bool = True
    ai_instructions: str = ""

    def to_json(self):
        return {"autoApprove": self.auto_approve, "chatMode": self.chat_mode,
                "enableAutocomplete": self.enable_autocomplete,
                "aiInstructions": self.ai_instructions}


class SettingsService:
    def __init__(self, storage: Optional[MemoryStorage] = None) -> None:
        self._storage = storage if storage is not None else MemoryStorage()
        self.global_settings = GlobalSettings()
        self.providers: Dict[str, Dict[str, Any]] = {
            "local": {"models": ["llama-3-8b", "llama-3-70b", "mixtral-8x7b"],
                      "enabled": True},
        }
        self.model_selection_of_feature: Dict[str, Optional[ModelSelection]] = {
            f: ModelSelection("local", "llama-3-8b") for f in FEATURES
        }
        self._listeners: List = []
        self._load()

    def on_did_change(self, fn) -> None:
        self._listeners.append(fn)

    def _fire(self) -> None:
        for fn in self._listeners:
            try:
                fn()
            except Exception:
                pass

    def set_chat_mode(self, mode: str) -> None:
        if mode not in CHAT_MODES:
            raise ValueError(f"unknown chat mode {mode!r}")
        self.global_settings.chat_mode = mode
        self._save()
        self._fire()

    def set_model_selection(self, feature: str, provider: str, model: str) -> None:
        if feature not in FEATURES:
            raise ValueError(f"unknown feature {feature!r}")
        self.model_selection_of_feature[feature] = ModelSelection(provider, model)
        self._save()
        self._fire()

    def set_auto_approve(self, approval_class: str, value: bool) -> None:
        self.global_settings.auto_approve[approval_class] = value
        self._save()
        self._fire()

    def _save(self) -> None:
        self._storage.store(SETTINGS_KEY, js_stringify({
            "globalSettings": self.global_settings.to_json(),
            "modelSelectionOfFeature": {
                f: (s.to_json() if s else None)
                for f, s in self.model_selection_of_feature.items()
            },
        }))
        if hasattr(self._storage, "flush"):
            self._storage.flush()

    def _load(self) -> None:
        raw = self._storage.get(SETTINGS_KEY)
        if not raw:
            return
        try:
            data = js_parse(raw)
            g = data.get("globalSettings", {})
            self.global_settings = GlobalSettings(
                auto_approve=g.get("autoApprove", {}),
                chat_mode=g.get("chatMode", "agent"),
                enable_autocomplete=g.get("enableAutocomplete", True),
                ai_instructions=g.get("aiInstructions", ""),
            )
            for f, s in (data.get("modelSelectionOfFeature") or {}).items():
                if f in FEATURES and s:
                    self.model_selection_of_feature[f] = ModelSelection(
                        s["providerName"], s["modelName"])
        except Exception:
            pass
