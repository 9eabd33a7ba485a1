"""Function registries for config resolution.

The reference delegates registries to spaCy/Thinc (``registry.resolve`` at
/root/reference/spacy_ray/worker.py:93); functions are referenced from configs
as ``@architectures = "name.v1"`` etc.  This is a from-scratch implementation
of the same contract: named registries, decorator registration, and lookup.
"""
from __future__ import annotations

import importlib
from typing import Callable, Dict


class Registry:
    def __init__(self, name: str) -> None:
        self.name = name
        self._store: Dict[str, Callable] = {}

    def register(self, name: str, func: Callable | None = None):
        if func is not None:
            self._store[name] = func
            return func

        def wrapper(f: Callable) -> Callable:
            self._store[name] = f
            return f

        return wrapper

    def __call__(self, name: str, func: Callable | None = None):
        return self.register(name, func)

    def get(self, name: str) -> Callable:
        if name not in self._store:
            raise KeyError(
                f"Can't find '{name}' in registry '{self.name}'. "
                f"Available: {sorted(self._store)[:50]}"
            )
        return self._store[name]

    def has(self, name: str) -> bool:
        return name in self._store

    def get_all(self) -> Dict[str, Callable]:
        return dict(self._store)


class registry:
    """Namespace of all registries (mirrors thinc/spaCy registry names that
    appear in training configs)."""

    architectures = Registry("architectures")
    optimizers = Registry("optimizers")
    schedules = Registry("schedules")
    batchers = Registry("batchers")
    loggers = Registry("loggers")
    readers = Registry("readers")
    tokenizers = Registry("tokenizers")
    factories = Registry("factories")
    callbacks = Registry("callbacks")
    scorers = Registry("scorers")
    augmenters = Registry("augmenters")
    initializers = Registry("initializers")
    misc = Registry("misc")

    _populated = False

    @classmethod
    def get_registry(cls, name: str) -> Registry:
        reg = getattr(cls, name, None)
        if not isinstance(reg, Registry):
            raise KeyError(f"Unknown registry '@{name}'")
        return reg

    @classmethod
    def ensure_populated(cls) -> None:
        """Import the modules whose import side-effects register the built-in
        functions (architectures, optimizers, ...).  Mirrors how the reference
        relies on spaCy's import-time registration."""
        if cls._populated:
            return
        cls._populated = True
        for mod in (
            "spacy_ray_amd.models.architectures",
            "spacy_ray_amd.train.optimizer",
            "spacy_ray_amd.train.schedules",
            "spacy_ray_amd.data.batcher",
            "spacy_ray_amd.data.corpus",
            "spacy_ray_amd.train.loggers",
            "spacy_ray_amd.pipeline.factories",
        ):
            importlib.import_module(mod)


def resolve_registry_ref(reg_name: str, func_name: str) -> Callable:
    registry.ensure_populated()
    return registry.get_registry(reg_name).get(func_name)
