"""LR scheduler registry (parity: reference
unicore/optim/lr_scheduler/__init__.py)."""

import importlib
import os

from unicore_amd import registry
from unicore_amd.optim.lr_scheduler.unicore_lr_scheduler import (  # noqa
    UnicoreLRScheduler,
)

(
    build_lr_scheduler_,
    register_lr_scheduler,
    LR_SCHEDULER_REGISTRY,
) = registry.setup_registry(
    "--lr-scheduler", base_class=UnicoreLRScheduler, default="fixed"
)


def build_lr_scheduler(args, optimizer, total_train_steps):
    return build_lr_scheduler_(args, optimizer, total_train_steps)


# automatically import any Python files in the lr_scheduler/ directory
for file in sorted(os.listdir(os.path.dirname(__file__))):
    if file.endswith(".py") and not file.startswith("_"):
        file_name = file[: file.find(".py")]
        importlib.import_module("unicore_amd.optim.lr_scheduler." + file_name)
