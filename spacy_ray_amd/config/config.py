"""spaCy/confection-compatible ``config.cfg`` parsing, interpolation and
resolution.

The reference leans on thinc's ``Config`` (`/root/reference/spacy_ray/
train_cli.py:44-46` ``load_config``; `/root/reference/spacy_ray/worker.py:
91-95` re-resolution per worker).  This module re-implements that contract
from scratch:

* INI-style files with dotted nested sections (``[training.optimizer]``),
* JSON-typed values (numbers, bools, null, strings, lists, inline dicts),
* ``${section.key}`` variable interpolation (whole-value refs keep type),
* ``@registryname = "func.v1"`` blocks resolved bottom-up via the registry,
* dotted CLI overrides (``--training.max_steps 100``).

Only the semantics the reference's configs exercise are implemented
(key list: SURVEY.md §5.6).
"""
from __future__ import annotations

import copy
import json
import re
from pathlib import Path
from typing import Any, Dict, Iterator, List, Optional, Tuple, Union

from .registry import registry as _registry

_SECTION_RE = re.compile(r"^\[([A-Za-z0-9_.*]+)\]\s*$")
_VAR_RE = re.compile(r"\$\{([A-Za-z0-9_.:]+)\}")


def _parse_value(raw: str) -> Any:
    raw = raw.strip()
    if raw == "":
        return ""
    try:
        return json.loads(raw)
    except Exception:
        return raw  # bare string


def _format_value(value: Any) -> str:
    if isinstance(value, str):
        # keep ${...} refs unquoted-looking but JSON-quote everything for
        # round-trip fidelity
        return json.dumps(value)
    return json.dumps(value)


class Config(dict):
    """A nested dict representing a parsed config file."""

    def __init__(self, data: Optional[Dict[str, Any]] = None) -> None:
        super().__init__()
        if data:
            self.update(copy.deepcopy(dict(data)))

    # ------------------------------------------------------------------ IO
    @classmethod
    def from_str(cls, text: str, *, overrides: Optional[Dict[str, Any]] = None,
                 interpolate: bool = False) -> "Config":
        cfg = cls()
        section: Optional[Dict[str, Any]] = None
        for lineno, line in enumerate(text.splitlines(), 1):
            stripped = line.strip()
            if not stripped or stripped.startswith(("#", ";")):
                continue
            m = _SECTION_RE.match(stripped)
            if m:
                section = cfg._make_section(m.group(1))
                continue
            if "=" not in stripped:
                raise ValueError(f"config parse error on line {lineno}: {line!r}")
            key, _, raw = stripped.partition("=")
            key = key.strip()
            value = _parse_value(raw)
            if section is None:
                cfg[key] = value
            else:
                section[key] = value
        if overrides:
            cfg.apply_overrides(overrides)
        if interpolate:
            return cfg.interpolate()
        return cfg

    @classmethod
    def from_disk(cls, path: Union[str, Path], *, overrides=None,
                  interpolate: bool = False) -> "Config":
        text = Path(path).read_text(encoding="utf8")
        return cls.from_str(text, overrides=overrides, interpolate=interpolate)

    def to_str(self) -> str:
        lines: List[str] = []
        flat_root = {k: v for k, v in self.items() if not isinstance(v, dict)}
        for k, v in flat_root.items():
            lines.append(f"{k} = {_format_value(v)}")
        for path, section in self._walk_sections():
            lines.append("")
            lines.append(f"[{path}]")
            for k, v in section.items():
                if not isinstance(v, dict):
                    lines.append(f"{k} = {_format_value(v)}")
        return "\n".join(lines) + "\n"

    def to_disk(self, path: Union[str, Path]) -> None:
        Path(path).write_text(self.to_str(), encoding="utf8")

    def copy(self) -> "Config":
        return Config(self)

    # ------------------------------------------------------ section helpers
    def _make_section(self, dotted: str) -> Dict[str, Any]:
        node: Dict[str, Any] = self
        for part in dotted.split("."):
            nxt = node.get(part)
            if not isinstance(nxt, dict):
                nxt = {}
                node[part] = nxt
            node = nxt
        return node

    def _walk_sections(self) -> Iterator[Tuple[str, Dict[str, Any]]]:
        def walk(prefix: str, node: Dict[str, Any]):
            for k, v in node.items():
                if isinstance(v, dict):
                    path = f"{prefix}.{k}" if prefix else k
                    yield path, v
                    yield from walk(path, v)

        yield from walk("", self)

    def get_dotted(self, dotted: str) -> Any:
        node: Any = self
        for part in dotted.split("."):
            if not isinstance(node, dict) or part not in node:
                raise KeyError(dotted)
            node = node[part]
        return node

    def set_dotted(self, dotted: str, value: Any) -> None:
        parts = dotted.split(".")
        node = self._make_section(".".join(parts[:-1])) if len(parts) > 1 else self
        node[parts[-1]] = value

    def apply_overrides(self, overrides: Dict[str, Any]) -> None:
        for dotted, value in overrides.items():
            if isinstance(value, str):
                value = _parse_value(value)
            self.set_dotted(dotted, value)

    # --------------------------------------------------------- interpolation
    def interpolate(self) -> "Config":
        """Substitute ``${a.b.c}`` refs.  A value that is exactly one ref
        keeps the referenced type; embedded refs do string substitution."""
        out = Config(self)

        def subst(value: Any, depth: int = 0) -> Any:
            if depth > 16:
                raise ValueError("circular config interpolation")
            if isinstance(value, str):
                m = _VAR_RE.fullmatch(value.strip())
                if m:
                    ref = m.group(1).replace(":", ".")
                    return subst(out.get_dotted(ref), depth + 1)
                return _VAR_RE.sub(
                    lambda mm: str(subst(out.get_dotted(mm.group(1).replace(":", ".")), depth + 1)),
                    value,
                )
            if isinstance(value, dict):
                return {k: subst(v, depth) for k, v in value.items()}
            if isinstance(value, list):
                return [subst(v, depth) for v in value]
            return value

        data = {k: subst(v) for k, v in out.items()}
        return Config(data)


def resolve(section: Dict[str, Any], *, schema=None, validate: bool = True) -> Any:
    """Resolve a config (sub)tree: build registered functions bottom-up.

    A dict containing an ``@<registry>`` key becomes the return value of the
    registered function called with the remaining keys as kwargs (nested
    sections resolved first) — the contract of thinc's ``registry.resolve``
    used at `/root/reference/spacy_ray/worker.py:93`.
    """
    _registry.ensure_populated()

    def _resolve_node(node: Any) -> Any:
        if isinstance(node, dict):
            at_keys = [k for k in node if k.startswith("@")]
            resolved = {k: _resolve_node(v) for k, v in node.items() if not k.startswith("@")}
            if at_keys:
                if len(at_keys) > 1:
                    raise ValueError(f"multiple @keys in one block: {at_keys}")
                reg_name = at_keys[0][1:]
                func_name = node[at_keys[0]]
                func = _registry.get_registry(reg_name).get(func_name)
                args = resolved.pop("*", None)
                if args is not None:
                    if isinstance(args, dict):
                        args = list(args.values())
                    return func(*args, **resolved)
                return func(**resolved)
            return resolved
        if isinstance(node, list):
            return [_resolve_node(v) for v in node]
        return node

    out = _resolve_node(dict(section))
    if schema is not None and validate and isinstance(out, dict):
        # pydantic-typed validation (spaCy's registry.resolve(schema=...)
        # contract); resolved callables/objects pass through unvalidated
        model = schema(**out)
        validated = {k: getattr(model, k) for k in type(model).model_fields}
        extra = model.model_extra or {}
        out = {**validated, **extra}
    return out


def resolve_dot_names(config: Config, dot_names: List[str]) -> Tuple[Any, ...]:
    """Resolve values referenced by dotted name into the config, building any
    ``@``-blocks found there (mirror of spaCy ``resolve_dot_names`` used at
    `/root/reference/spacy_ray/worker.py:94-95`)."""
    out = []
    for name in dot_names:
        node = config.get_dotted(name)
        out.append(resolve(node) if isinstance(node, dict) else node)
    return tuple(out)


def parse_config_overrides(args: List[str]) -> Dict[str, Any]:
    """Parse trailing CLI args of the form ``--a.b.c value`` into an override
    dict (contract of spaCy's ``parse_config_overrides`` consumed at
    `/root/reference/spacy_ray/train_cli.py:44`)."""
    overrides: Dict[str, Any] = {}
    i = 0
    while i < len(args):
        arg = args[i]
        if not arg.startswith("--"):
            raise ValueError(f"unexpected override argument: {arg}")
        key = arg[2:]
        if "=" in key:
            key, _, val = key.partition("=")
            overrides[key] = _parse_value(val)
            i += 1
        else:
            if i + 1 >= len(args):
                raise ValueError(f"override {arg} missing a value")
            overrides[key] = _parse_value(args[i + 1])
            i += 2
    return overrides
