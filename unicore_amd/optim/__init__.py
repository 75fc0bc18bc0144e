"""Optimizer registry (parity: reference unicore/optim/__init__.py)."""

import importlib
import os

from unicore_amd import registry
from unicore_amd.optim.unicore_optimizer import UnicoreOptimizer  # noqa
from unicore_amd.optim.fp16_optimizer import (  # noqa
    FP16Optimizer,
    separate_decay_params,
)

__all__ = [
    "UnicoreOptimizer",
    "FP16Optimizer",
]

(_build_optimizer, register_optimizer, OPTIMIZER_REGISTRY) = registry.setup_registry(
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


# automatically import any Python files in the optim/ directory
for file in sorted(os.listdir(os.path.dirname(__file__))):
    if file.endswith(".py") and not file.startswith("_"):
        file_name = file[: file.find(".py")]
        importlib.import_module("unicore_amd.optim." + file_name)

from unicore_amd.optim import lr_scheduler  # noqa
