"""Loss registry (parity: reference unicore/losses/__init__.py)."""

import importlib
import os

from unicore_amd import registry
from unicore_amd.losses.unicore_loss import UnicoreLoss

build_loss_, register_loss, LOSS_REGISTRY = registry.setup_registry(
    "--loss", base_class=UnicoreLoss, default="cross_entropy"
)


def build_loss(args, task):
    return build_loss_(args, task)


# automatically import any Python files in the losses/ directory
for file in sorted(os.listdir(os.path.dirname(__file__))):
    if file.endswith(".py") and not file.startswith("_"):
        file_name = file[: file.find(".py")]
        if file_name != "unicore_loss":
            importlib.import_module("unicore_amd.losses." + file_name)
