"""Model base class (parity: reference unicore/models/unicore_model.py:18-58)."""

import logging
from argparse import Namespace

import torch.nn as nn

logger = logging.getLogger(__name__)


class BaseUnicoreModel(nn.Module):
    """Base class for models."""

    def __init__(self):
        super().__init__()

    @classmethod
    def add_args(cls, parser):
        """Add model-specific arguments to the parser."""
        pass

    @classmethod
    def build_model(cls, args, task):
        """Build a new model instance."""
        raise NotImplementedError("Model must implement the build_model method")

    def extract_features(self, *args, **kwargs):
        """Similar to *forward* but only return features."""
        return self(*args, **kwargs)

    def load_state_dict(
        self,
        state_dict,
        strict=True,
        model_args: Namespace = None,
    ):
        """Copies parameters and buffers from *state_dict* into this module and
        its descendants.

        Overrides the method in :class:`nn.Module`. Compared with that method
        this additionally accepts *model_args*, giving models the chance to
        adapt checkpoints (e.g. resize heads) before loading.
        """
        return super().load_state_dict(state_dict, strict)

    def set_num_updates(self, num_updates):
        """State from trainer to pass along to model at every update."""

        def _apply(m):
            if hasattr(m, "set_num_updates") and m != self:
                m.set_num_updates(num_updates)

        self.apply(_apply)
