"""YAML app/template loader (reference internals/yaml_loader.py).

Supports the reference's template syntax:
  $ref-style object instantiation:   !pw.xpacks.llm.llms.EchoChat
  variables:                         $llm: !pw...   then  llm: $llm
"""
from __future__ import annotations

import importlib
from typing import Any

import yaml


def _resolve_callable(tag: str):
    path = tag.lstrip("!")
    if path.startswith("pw."):
        path = "pathway_amd." + path[3:]
    module_path, _, attr = path.rpartition(".")
    mod = importlib.import_module(module_path)
    return getattr(mod, attr)


class _Ctor:
    def __init__(self, fn, kwargs):
        self.fn = fn
        self.kwargs = kwargs

    def build(self, variables):
        kwargs = {
            k: _materialize(v, variables) for k, v in (self.kwargs or {}).items()
        }
        return self.fn(**kwargs)


def _materialize(v, variables):
    if isinstance(v, _Ctor):
        return v.build(variables)
    if isinstance(v, str) and v.startswith("$") and v[1:] in variables:
        return _materialize(variables[v[1:]], variables)
    if isinstance(v, list):
        return [_materialize(x, variables) for x in v]
    if isinstance(v, dict):
        return {k: _materialize(x, variables) for k, x in v.items()}
    return v


def load_yaml(stream) -> Any:
    class Loader(yaml.SafeLoader):
        pass

    def multi_ctor(loader, tag_suffix, node):
        fn = _resolve_callable(tag_suffix)
        if isinstance(node, yaml.MappingNode):
            kwargs = loader.construct_mapping(node, deep=True)
        else:
            kwargs = {}
        return _Ctor(fn, kwargs)

    Loader.add_multi_constructor("!", lambda l, s, n: multi_ctor(l, s, n))
    raw = yaml.load(stream, Loader=Loader)
    if not isinstance(raw, dict):
        return raw
    variables = {k[1:]: v for k, v in raw.items() if isinstance(k, str) and k.startswith("$")}
    out = {}
    for k, v in raw.items():
        if isinstance(k, str) and k.startswith("$"):
            continue
        out[k] = _materialize(v, variables)
    return out
