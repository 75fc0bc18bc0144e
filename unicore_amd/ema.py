"""Exponential moving average of model weights.

Parity: reference unicore/ema.py:6-65 — a deep-copied fp32 shadow model.
When training in fp16/bf16 the shadow is flattened per decay group with the
SAME (group, dtype, padding) layout as FP16Optimizer's fp32 masters, so
each update is one fused lerp per group fed directly from
``optimizer.fp32_params`` (reference unicore/trainer.py:720-725).
Serialized as {"params": state_dict, "decay": float}.
"""

from copy import deepcopy

import torch


class ExponentialMovingAverageModel:
    def __init__(self, args, model, decay, is_flattened=False):
        self.args = args
        self.model_ema = deepcopy(model)
        self.decay = decay
        self.is_flattened = is_flattened
        if is_flattened:
            self.flatten_params = self._flatten_like_optimizer()
        else:
            self.name2param = self._promote_params()

    def _promote_params(self):
        """Per-name fp32 views of the shadow model (slow path)."""
        table = {}
        for name, param in self.model_ema.named_parameters():
            table[name] = param
            param.data = param.data.float()  # EMA state is always fp32
            param.grad = None
        return table

    def _flatten_like_optimizer(self):
        """Flatten the shadow with FP16Optimizer's exact group layout so
        update() can consume optimizer.fp32_params positionally."""
        from unicore_amd.optim.fp16_optimizer import (
            flatten_fp32_master,
            separate_decay_params,
        )

        groups = separate_decay_params(
            self.args, self.model_ema.named_parameters()
        )
        return [
            flatten_fp32_master(g["params"], set_to_param=True) for g in groups
        ]

    @torch.no_grad()
    def _lerp(self, shadow, live):
        # shadow -= (1 - decay) * (shadow - live)
        shadow.add_(shadow - live.data.float(), alpha=self.decay - 1.0)

    def update(self, new_param):
        """Blend the live weights in. *new_param* is either the optimizer's
        flat fp32 masters (flattened mode) or model.named_parameters()."""
        with torch.no_grad():
            if self.is_flattened:
                for shadow, live in zip(self.flatten_params, new_param):
                    self._lerp(shadow.data, live)
            else:
                for name, live in new_param:
                    if name in self.name2param:
                        self._lerp(self.name2param[name].data, live)

    def state_dict(self) -> dict:
        return {"params": self.model_ema.state_dict(), "decay": self.decay}

    def load_state_dict(self, state: dict) -> None:
        self.model_ema.load_state_dict(state["params"])
        self.decay = state.get("decay", self.decay)
