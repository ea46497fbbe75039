"""Global component registries + metadata.

Mirrors the reference's per-kind ``lazy_static RwLock<HashMap<String,
Arc<dyn XxxBuilder>>>`` registries keyed by the YAML ``type:`` string
(input/mod.rs:141 et al.) and the component metadata registry with per-type
JSON Schema + example used for discovery and validation
(component/mod.rs:36-167).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .errors import ConfigError

KINDS = ("input", "output", "processor", "buffer", "codec", "temporary",
         "wal_store")


@dataclass
class ComponentMetadata:
    kind: str
    name: str
    description: str = ""
    config_schema: Optional[dict] = None   # JSON Schema for the flattened blob
    example: Optional[dict] = None
    tags: List[str] = field(default_factory=list)


class _Registry:
    def __init__(self, kind: str):
        self.kind = kind
        self.builders: Dict[str, Callable] = {}
        self.metadata: Dict[str, ComponentMetadata] = {}

    def register(self, name: str, builder: Callable,
                 metadata: Optional[ComponentMetadata] = None) -> None:
        if name in self.builders:
            raise ConfigError(f"{self.kind} builder {name!r} already registered")
        self.builders[name] = builder
        self.metadata[name] = metadata or ComponentMetadata(self.kind, name)

    def build(self, type_name: str, config: dict, resource=None):
        b = self.builders.get(type_name)
        if b is None:
            raise ConfigError(
                f"unknown {self.kind} type {type_name!r}; "
                f"registered: {sorted(self.builders)}"
            )
        return b(config or {}, resource)

    def names(self) -> List[str]:
        return sorted(self.builders)


_REGISTRIES: Dict[str, _Registry] = {k: _Registry(k) for k in KINDS}


def registry(kind: str) -> _Registry:
    try:
        return _REGISTRIES[kind]
    except KeyError:
        raise ConfigError(f"unknown component kind {kind!r}") from None


def _json_type(v) -> dict:
    if isinstance(v, bool):
        return {"type": "boolean"}
    if isinstance(v, int):
        return {"type": "integer"}
    if isinstance(v, float):
        return {"type": "number"}
    if isinstance(v, str):
        return {"type": "string"}
    if isinstance(v, list):
        item = _json_type(v[0]) if v else {}
        return {"type": "array", "items": item}
    if isinstance(v, dict):
        return {"type": "object",
                "properties": {k: _json_type(x) for k, x in v.items()}}
    return {}


def schema_from_example(example: dict) -> dict:
    """Minimal JSON Schema derived from a component's example config
    (reference components publish hand-written schemas,
    component/mod.rs:96; the derived form covers discovery + docs and is
    refined per component over time)."""
    props = {k: _json_type(v) for k, v in example.items() if k != "type"}
    props["type"] = {"type": "string", "const": example.get("type", "")}
    return {"type": "object", "properties": props,
            "additionalProperties": True}


def register(kind: str, name: str, *, description: str = "",
             config_schema: Optional[dict] = None, example: Optional[dict] = None):
    """Decorator: ``@register("input", "generate")`` on a builder callable."""
    def deco(builder):
        schema = config_schema
        if schema is None and example:
            schema = schema_from_example({**example, "type": name})
        registry(kind).register(
            name, builder,
            ComponentMetadata(kind, name, description, schema, example),
        )
        return builder
    return deco


def build_component(kind: str, spec: dict, resource=None):
    """Build from a flattened YAML blob ``{type: name, ...rest}``
    (reference XxxConfig with #[serde(flatten)])."""
    if not isinstance(spec, dict) or "type" not in spec:
        raise ConfigError(f"{kind} spec must be a mapping with a 'type' key: {spec!r}")
    spec = dict(spec)
    type_name = spec.pop("type")
    return registry(kind).build(type_name, spec, resource)


def list_components(kind: Optional[str] = None) -> List[ComponentMetadata]:
    kinds = [kind] if kind else list(KINDS)
    out: List[ComponentMetadata] = []
    for k in kinds:
        out.extend(_REGISTRIES[k].metadata[n] for n in _REGISTRIES[k].names())
    return out


def component_metadata(kind: str, name: str) -> ComponentMetadata:
    md = _REGISTRIES[kind].metadata.get(name)
    if md is None:
        raise ConfigError(f"unknown {kind} {name!r}")
    return md


def build_config_schema() -> dict:
    """Full engine JSON schema for ``arkflow schema``
    (reference component/mod.rs build_config_schema)."""
    def kind_schema(kind):
        return {
            "type": "object",
            "required": ["type"],
            "properties": {
                "type": {"enum": registry(kind).names()},
            },
        }

    return {
        "$schema": "https://json-schema.org/draft/2020-12/schema",
        "title": "arkflow_amd engine configuration",
        "type": "object",
        "required": ["streams"],
        "properties": {
            "streams": {
                "type": "array",
                "items": {
                    "type": "object",
                    "required": ["input", "pipeline", "output"],
                    "properties": {
                        "input": kind_schema("input"),
                        "output": kind_schema("output"),
                        "error_output": kind_schema("output"),
                        "buffer": kind_schema("buffer"),
                        "pipeline": {
                            "type": "object",
                            "properties": {
                                "thread_num": {"type": "integer"},
                                "processors": {
                                    "type": "array",
                                    "items": kind_schema("processor"),
                                },
                            },
                        },
                    },
                },
            },
        },
    }
