"""Model base class (parity: reference unicore/models/unicore_model.py:18-58)."""

import logging
from argparse import Namespace

import torch.nn as nn

logger = logging.getLogger(__name__)


class BaseUnicoreModel(nn.Module):
    """Base for all registered models: built by ``build_model(args, task)``,
    checkpoint-loaded with an optional args namespace for shape adaptation,
    and notified of the update count every step."""

    def __init__(self):
        super().__init__()

    @classmethod
    def add_args(cls, parser):
        """Hook for model-specific CLI arguments."""

    @classmethod
    def build_model(cls, args, task):
        """Construct an instance from the parsed args + task."""
        raise NotImplementedError(
            f"{cls.__name__} must implement build_model"
        )

    def extract_features(self, *args, **kwargs):
        """Forward returning features only (defaults to plain forward)."""
        return self(*args, **kwargs)

    def load_state_dict(self, state_dict, strict=True,
                        model_args: Namespace = None):
        """nn.Module.load_state_dict plus a *model_args* hook: subclasses
        may inspect it to adapt checkpoints (e.g. resize heads) before the
        actual load."""
        return super().load_state_dict(state_dict, strict)

    def set_num_updates(self, num_updates):
        """Propagate the trainer's update count to every submodule that
        cares (e.g. schedules inside the model)."""

        def _notify(m):
            if m is not self and hasattr(m, "set_num_updates"):
                m.set_num_updates(num_updates)

        self.apply(_notify)
