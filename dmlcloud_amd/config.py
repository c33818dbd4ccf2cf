"""Dependency-free experiment configuration.

The reference delegates its config store to OmegaConf
(reference dmlcloud/pipeline.py:21-27, checkpoint.py:105-117). This
framework ships its own minimal equivalent so the library has no
third-party config dependency: nested dot-access mapping, YAML
round-trip, and ``${a.b}``-style interpolation on resolve.

Public surface used by the pipeline:
    Config.create(obj)      -> Config       (dict | Config | None)
    cfg.a.b / cfg['a']['b'] -> values
    cfg.to_container(resolve=True) -> plain dict
    cfg.to_yaml(resolve=True)      -> str
    Config.load(path) / cfg.save(path)
"""

from __future__ import annotations

import copy
import re
from pathlib import Path
from typing import Any, Dict, Optional, Union

import yaml

# \w covers unicode identifiers too (any key YAML can express and Python
# can attribute-access); dots path into nested configs
_INTERP_RE = re.compile(r'\$\{([\w.]+)\}')


class Config:
    """A nested, dot-accessible configuration mapping."""

    def __init__(self, data: Optional[Dict[str, Any]] = None):
        object.__setattr__(self, '_data', {})
        if data:
            for key, value in data.items():
                self[key] = value

    # ---------------------------------------------------------- construction

    @staticmethod
    def create(obj: Union['Config', Dict, None] = None) -> 'Config':
        if obj is None:
            return Config()
        if isinstance(obj, Config):
            return obj
        if isinstance(obj, dict):
            return Config(obj)
        raise ValueError(f'Cannot create Config from {type(obj)}')

    @staticmethod
    def load(path: Union[str, Path]) -> 'Config':
        with open(path) as f:
            data = yaml.safe_load(f)
        return Config.create(data or {})

    # ---------------------------------------------------------- mapping API

    def __getitem__(self, key: str) -> Any:
        return self._data[key]

    def __setitem__(self, key: str, value: Any):
        if isinstance(value, dict):
            value = Config(value)
        self._data[key] = value

    def __delitem__(self, key: str):
        del self._data[key]

    def __contains__(self, key: str) -> bool:
        return key in self._data

    def __iter__(self):
        return iter(self._data)

    def __len__(self) -> int:
        return len(self._data)

    def keys(self):
        return self._data.keys()

    def values(self):
        return self._data.values()

    def items(self):
        return self._data.items()

    def get(self, key: str, default: Any = None) -> Any:
        return self._data.get(key, default)

    def setdefault(self, key: str, default: Any = None) -> Any:
        if key not in self:
            self[key] = default
        return self[key]

    # -------------------------------------------------------- attribute API

    def __getattr__(self, name: str) -> Any:
        if name.startswith('_'):
            raise AttributeError(name)
        try:
            return self._data[name]
        except KeyError:
            raise AttributeError(f'Config has no key {name!r}') from None

    def __setattr__(self, name: str, value: Any):
        if name.startswith('_'):
            object.__setattr__(self, name, value)
        else:
            self[name] = value

    # -------------------------------------------------------- serialization

    def to_container(self, resolve: bool = False) -> Dict[str, Any]:
        """Convert to a plain nested dict. If resolve, apply interpolation."""
        container = self._as_dict()
        if resolve:
            container = _resolve_container(container)
        return container

    def _as_dict(self) -> Dict[str, Any]:
        out = {}
        for key, value in self._data.items():
            if isinstance(value, Config):
                out[key] = value._as_dict()
            else:
                out[key] = copy.deepcopy(value)
        return out

    def to_yaml(self, resolve: bool = False) -> str:
        return yaml.safe_dump(self.to_container(resolve=resolve), sort_keys=False, default_flow_style=False)

    def save(self, path: Union[str, Path], resolve: bool = False):
        with open(path, 'w') as f:
            f.write(self.to_yaml(resolve=resolve))

    def merge(self, other: Union['Config', Dict]) -> 'Config':
        """Deep-merge `other` into a copy of self (other wins)."""
        merged = Config(self._as_dict())
        other = Config.create(other if not isinstance(other, Config) else other._as_dict())
        for key, value in other.items():
            if key in merged and isinstance(merged[key], Config) and isinstance(value, Config):
                merged[key] = merged[key].merge(value)
            else:
                merged[key] = value
        return merged

    def __eq__(self, other) -> bool:
        if isinstance(other, Config):
            return self._as_dict() == other._as_dict()
        if isinstance(other, dict):
            return self._as_dict() == other
        return NotImplemented

    def __repr__(self) -> str:
        return f'Config({self._as_dict()!r})'


def _lookup(container: Dict[str, Any], dotted: str) -> Any:
    node: Any = container
    for part in dotted.split('.'):
        if not isinstance(node, dict) or part not in node:
            raise KeyError(f'Interpolation key {dotted!r} not found')
        node = node[part]
    return node


def _resolve_value(value: Any, root: Dict[str, Any], depth: int = 0) -> Any:
    if depth > 16:
        raise ValueError('Interpolation recursion limit exceeded (cycle?)')
    if isinstance(value, str):
        full = _INTERP_RE.fullmatch(value)
        if full:  # whole-string interpolation preserves the referenced type
            return _resolve_value(_lookup(root, full.group(1)), root, depth + 1)
        return _INTERP_RE.sub(lambda m: str(_resolve_value(_lookup(root, m.group(1)), root, depth + 1)), value)
    if isinstance(value, dict):
        return {k: _resolve_value(v, root, depth) for k, v in value.items()}
    if isinstance(value, list):
        return [_resolve_value(v, root, depth) for v in value]
    return value


def _resolve_container(container: Dict[str, Any]) -> Dict[str, Any]:
    return {k: _resolve_value(v, container) for k, v in container.items()}
