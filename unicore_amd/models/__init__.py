"""Model + architecture registries (parity: reference
unicore/models/__init__.py:17-102).

Two levels: MODEL_REGISTRY maps a model family name to its class;
ARCH_MODEL_REGISTRY maps each ``--arch`` choice to that class, with
ARCH_CONFIG_REGISTRY holding the per-arch default-filling function applied
after parsing. Modules in this directory import below so built-in models
self-register.
"""

import importlib
import os

from .distributed_unicore_model import DistributedUnicoreModel  # noqa: F401
from .unicore_model import BaseUnicoreModel

MODEL_REGISTRY = {}
ARCH_MODEL_REGISTRY = {}
ARCH_MODEL_INV_REGISTRY = {}
ARCH_CONFIG_REGISTRY = {}

__all__ = [
    "BaseUnicoreModel",
    "DistributedUnicoreModel",
    "register_model",
    "register_model_architecture",
    "build_model",
]


def build_model(args, task):
    return ARCH_MODEL_REGISTRY[args.arch].build_model(args, task)


def register_model(name):
    """Decorator registering a BaseUnicoreModel subclass under *name*."""

    def wrap(cls):
        if name in MODEL_REGISTRY:
            raise ValueError(f"Cannot register duplicate model ({name})")
        if not issubclass(cls, BaseUnicoreModel):
            raise ValueError(
                f"Model ({name}: {cls.__name__}) must extend BaseUnicoreModel"
            )
        MODEL_REGISTRY[name] = cls
        return cls

    return wrap


def register_model_architecture(model_name, arch_name):
    """Decorator registering an ``--arch`` choice for an existing model:
    the decorated function fills arch-specific defaults into args after
    parsing."""

    def wrap(fn):
        if model_name not in MODEL_REGISTRY:
            raise ValueError(
                "Cannot register model architecture for unknown model type "
                f"({model_name})"
            )
        if arch_name in ARCH_MODEL_REGISTRY:
            raise ValueError(
                f"Cannot register duplicate model architecture ({arch_name})"
            )
        if not callable(fn):
            raise ValueError(
                f"Model architecture must be callable ({arch_name})"
            )
        ARCH_MODEL_REGISTRY[arch_name] = MODEL_REGISTRY[model_name]
        ARCH_MODEL_INV_REGISTRY.setdefault(model_name, []).append(arch_name)
        ARCH_CONFIG_REGISTRY[arch_name] = fn
        return fn

    return wrap


def _import_all_model_modules():
    here = os.path.dirname(__file__)
    skip = ("unicore_model", "distributed_unicore_model")
    for entry in sorted(os.listdir(here)):
        if entry.startswith(("_", ".")):
            continue
        is_pkg = os.path.isdir(os.path.join(here, entry))
        if not (entry.endswith(".py") or is_pkg):
            continue
        modname = entry[:-3] if entry.endswith(".py") else entry
        if modname not in skip:
            importlib.import_module(f"unicore_amd.models.{modname}")


_import_all_model_modules()
