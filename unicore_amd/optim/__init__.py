"""Optimizer registry (parity: reference unicore/optim/__init__.py).

``build_optimizer`` splits params into decay/no-decay groups by name
before handing them to the registered optimizer class; every module in
this directory imports below so built-in optimizers self-register.
"""

import importlib
import os

from unicore_amd import registry
from unicore_amd.optim.unicore_optimizer import UnicoreOptimizer  # noqa
from unicore_amd.optim.fp16_optimizer import (  # noqa
    FP16Optimizer,
    separate_decay_params,
)

__all__ = ["UnicoreOptimizer", "FP16Optimizer"]

(_build_optimizer, register_optimizer, OPTIMIZER_REGISTRY) = \
    registry.setup_registry(
        "--optimizer", base_class=UnicoreOptimizer, default="adam"
    )


def build_optimizer(args, params, separate=True, *extra_args, **extra_kwargs):
    """Build the registered optimizer over *params*.

    *params* is a list of (name, param) pairs when ``separate=True`` (the
    decay/no-decay split needs names); otherwise raw params/param-groups.
    """
    if separate:
        params = separate_decay_params(args, list(params))
    return _build_optimizer(args, params, *extra_args, **extra_kwargs)


def build_raw_optimizer(args, param_groups):
    """Build the registered optimizer over explicit param groups."""
    return _build_optimizer(args, param_groups)


def _import_all_optim_modules():
    here = os.path.dirname(__file__)
    for entry in sorted(os.listdir(here)):
        if entry.endswith(".py") and not entry.startswith("_"):
            importlib.import_module(f"unicore_amd.optim.{entry[:-3]}")


_import_all_optim_modules()

from unicore_amd.optim import lr_scheduler  # noqa
