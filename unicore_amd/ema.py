"""Exponential moving average of model weights.

Parity: reference unicore/ema.py:6-65 — a deep-copied fp32 EMA model; when
training in fp16/bf16 the EMA params are flattened per decay group (same
layout as the optimizer's fp32 master flats) so each update is one fused
lerp per group fed directly from ``optimizer.fp32_params``
(reference unicore/trainer.py:720-725). Checkpointable as
{"params": state_dict, "decay": float}.
"""

from copy import deepcopy

import torch


class ExponentialMovingAverageModel:
    def __init__(self, args, model, decay, is_flattened=False):
        self.args = args
        self.model_ema = deepcopy(model)
        self.decay = decay
        self.is_flattened = is_flattened
        if not is_flattened:
            self.name2param = self.get_name2param()
        else:
            self.flatten_params = self.flatten_parameters()

    def get_name2param(self):
        name2param = dict()
        for n, p in self.model_ema.named_parameters():
            name2param[n] = p
            # keep EMA state in fp32 regardless of model dtype
            p.data = p.data.float()
            p.grad = None
        return name2param

    def flatten_parameters(self):
        """Flatten the EMA copy with the same (decay-group, dtype, padding)
        layout as FP16Optimizer's fp32 master flats, so update() can consume
        optimizer.fp32_params directly."""
        from unicore_amd.optim.fp16_optimizer import (
            flatten_fp32_master,
            separate_decay_params,
        )

        param_group = separate_decay_params(self.args, self.model_ema.named_parameters())
        return [
            flatten_fp32_master(pd["params"], set_to_param=True)
            for pd in param_group
        ]

    @torch.no_grad()
    def _lerp(self, ema_param, new_param):
        # ema -= (1 - decay) * (ema - new)
        ema_param.add_(ema_param - new_param.data.float(), alpha=-(1.0 - self.decay))

    def update(self, new_param):
        if self.is_flattened:
            # new_param: optimizer.fp32_params (one flat per decay group)
            with torch.no_grad():
                for e, p in zip(self.flatten_params, new_param):
                    self._lerp(e.data, p)
        else:
            # new_param: model.named_parameters()
            with torch.no_grad():
                for n, p in new_param:
                    if n in self.name2param:
                        self._lerp(self.name2param[n].data, p)

    def load_state_dict(self, state_dict):
        self.model_ema.load_state_dict(state_dict["params"])
        self.decay = state_dict.get("decay", self.decay)

    def state_dict(self):
        return {
            "params": self.model_ema.state_dict(),
            "decay": self.decay,
        }
