"""Attr/state_dict passthrough for DDP-wrapped modules
(parity: reference unicore/distributed/module_proxy_wrapper.py:10-62).

A DDP engine wraps the user model one level down (``wrapper.module``);
this proxy keeps the outer object behaving like the inner model: unknown
attributes resolve inward, and (de)serialization targets the inner module
so checkpoints never contain DDP prefixes.
"""

from torch import nn


class ModuleProxyWrapper(nn.Module):
    """Usage::

        model.some_attr = 123
        wrapped = ModuleProxyWrapper(SomeDDPEngine(model, ...))
        assert wrapped.some_attr == 123
        assert wrapped.state_dict().keys() == model.state_dict().keys()
    """

    def __init__(self, module: nn.Module):
        super().__init__()
        if not hasattr(module, "module"):
            raise AssertionError(
                "ModuleProxyWrapper wraps a DDP engine that itself wraps "
                "the model"
            )
        self.module = module

    def __getattr__(self, name):
        # resolution order: nn.Module machinery -> the DDP engine -> the
        # user model inside it
        try:
            return super().__getattr__(name)
        except AttributeError:
            engine = self._modules["module"]
            if hasattr(engine, name):
                return getattr(engine, name)
            return getattr(engine.module, name)

    def state_dict(self, *args, **kwargs):
        """Serialize the INNER model (no DDP key prefixes)."""
        inner = self.module.module
        return inner.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        """Restore into the inner model."""
        inner = self.module.module
        return inner.load_state_dict(*args, **kwargs)

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    def half(self):
        self.module.half()
        return self

    def bfloat16(self):
        self.module.bfloat16()
        return self
