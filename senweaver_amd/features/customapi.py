"""CustomApiService — user-defined API tool registration.

Rebuild of common/customApiService.ts (216 LoC): CRUD over persisted
CustomApiDefinition records (storage key ``senweaver.customApis``) and the
prompt-facing ``get_api_list_description`` that advertises enabled APIs to
the assistant for use through the ``api_request`` tool (the description
format, including its Chinese field labels, is kept verbatim).  The actual
HTTP execution lives in the api_request tool: in this offline environment
only loopback targets can succeed; registration/persistence is fully
functional either way.
"""

from __future__ import annotations

import time
import uuid
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from ..storage import FileStorage, MemoryStorage
from ..utils.jsonutil import js_parse, js_stringify

CUSTOM_API_STORAGE_KEY = "senweaver.customApis"

_FIELD_KEYS = ("name", "type", "required", "description", "defaultValue")


@dataclass
class CustomApiField:
    name: str
    type: str = "string"     # string | number | boolean | object | array
    required: bool = False
    description: str = ""
    defaultValue: Optional[str] = None


@dataclass
class CustomApiDefinition:
    id: str
    name: str
    url: str
    method: str              # GET | POST | PUT | DELETE | PATCH
    description: str
    fields: List[CustomApiField] = field(default_factory=list)
    headers: Optional[Dict[str, str]] = None
    responseDescription: Optional[str] = None
    enabled: bool = True
    createdAt: int = 0
    updatedAt: int = 0

    def to_dict(self) -> dict:
        d = {"id": self.id, "name": self.name, "url": self.url,
             "method": self.method, "description": self.description,
             "fields": [{k: getattr(f, k) for k in _FIELD_KEYS
                         if getattr(f, k) is not None}
                        for f in self.fields],
             "enabled": self.enabled,
             "createdAt": self.createdAt, "updatedAt": self.updatedAt}
        if self.headers is not None:
            d["headers"] = self.headers
        if self.responseDescription is not None:
            d["responseDescription"] = self.responseDescription
        return d

    @classmethod
    def from_dict(cls, d: dict) -> "CustomApiDefinition":
        fields = [CustomApiField(**{k: f.get(k) for k in _FIELD_KEYS
                                    if f.get(k) is not None})
                  for f in d.get("fields", [])]
        return cls(id=d["id"], name=d.get("name", ""), url=d.get("url", ""),
                   method=d.get("method", "GET"),
                   description=d.get("description", ""), fields=fields,
                   headers=d.get("headers"),
                   responseDescription=d.get("responseDescription"),
                   enabled=bool(d.get("enabled", True)),
                   createdAt=d.get("createdAt", 0),
                   updatedAt=d.get("updatedAt", 0))


class CustomApiService:
    def __init__(self, storage=None,
                 clock: Optional[Callable[[], int]] = None) -> None:
        self._storage = storage if storage is not None else MemoryStorage()
        self._clock = clock or (lambda: int(time.time() * 1000))
        self._apis: List[CustomApiDefinition] = []
        self._listeners: List[Callable[[], None]] = []
        self._load()

    # ---- persistence ----
    def _load(self) -> None:
        raw = self._storage.get(CUSTOM_API_STORAGE_KEY)
        if not raw:
            return
        try:
            state = js_parse(raw)
            self._apis = [CustomApiDefinition.from_dict(a)
                          for a in state.get("apis", [])]
        except Exception:
            self._apis = []

    def _save(self) -> None:
        self._storage.store(CUSTOM_API_STORAGE_KEY,
                            js_stringify({"apis": [a.to_dict() for a in self._apis]}))
        for fn in self._listeners:
            fn()

    def on_did_change_state(self, fn: Callable[[], None]) -> None:
        self._listeners.append(fn)

    # ---- CRUD ----
    def add_api(self, name: str, url: str, method: str, description: str,
                fields: Optional[List[CustomApiField]] = None,
                headers: Optional[Dict[str, str]] = None,
                response_description: Optional[str] = None,
                enabled: bool = True) -> CustomApiDefinition:
        now = self._clock()
        api = CustomApiDefinition(
            id=f"api_{now}_{uuid.uuid4().hex[:9]}", name=name, url=url,
            method=method, description=description, fields=fields or [],
            headers=headers, responseDescription=response_description,
            enabled=enabled, createdAt=now, updatedAt=now)
        self._apis.append(api)
        self._save()
        return api

    def update_api(self, api_id: str, **updates) -> None:
        api = self.get_api(api_id)
        if api is None:
            raise KeyError(f"unknown custom API {api_id!r}")
        for k, v in updates.items():
            if k in ("id", "createdAt"):
                continue
            setattr(api, k, v)
        api.updatedAt = self._clock()
        self._save()

    def delete_api(self, api_id: str) -> None:
        self._apis = [a for a in self._apis if a.id != api_id]
        self._save()

    def get_api(self, api_id: str) -> Optional[CustomApiDefinition]:
        return next((a for a in self._apis if a.id == api_id), None)

    def get_enabled_apis(self) -> List[CustomApiDefinition]:
        return [a for a in self._apis if a.enabled]

    @property
    def state(self) -> dict:
        return {"apis": [a.to_dict() for a in self._apis]}

    # ---- prompt surface (format verbatim from the reference) ----
    def get_api_list_description(self) -> str:
        enabled = self.get_enabled_apis()
        if not enabled:
            return ""
        parts = []
        for api in enabled:
            fields_desc = "\n".join(
                f"  - {f.name} ({f.type}{', 必填' if f.required else ''}): "
                f"{f.description}" for f in api.fields)
            resp = (f"- 响应说明: {api.responseDescription}"
                    if api.responseDescription else "")
            parts.append(f"## {api.name}\n- URL: {api.url}\n- 方法: {api.method}\n"
                         f"- 描述: {api.description}\n- 字段:\n{fields_desc}\n{resp}")
        body = "\n\n".join(parts)
        return (f"# 可用的自定义 API 列表\n\n以下 API 可以通过 api_request 工具调用：\n\n"
                f"{body}\n\n调用示例：使用 api_request 工具，设置对应的 url、method、"
                "headers 和 body 参数。")
