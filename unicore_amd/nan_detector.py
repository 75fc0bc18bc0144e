"""NaN/Inf forensics: module fwd/bwd hooks that report the first bad tensor.

Parity: reference unicore/nan_detector.py:15-109 — the trainer wraps a
re-run of the failing batch in this context after a FloatingPointError
(reference unicore/trainer.py:727-748); on exit it also dumps every
non-finite grad norm.
"""

import logging

import torch

logger = logging.getLogger(__name__)


class NanDetector:
    """Context manager that hooks every submodule and warns (once per
    direction) on the first NaN/Inf seen in a forward output or backward
    grad, naming the module that produced it."""

    def __init__(self, model, forward=True, backward=True):
        self.fhooks, self.bhooks = [], []
        self.forward, self.backward = forward, backward
        self.named_parameters = [*model.named_parameters()]
        self.reset()
        for name, submodule in model.named_modules():
            submodule.__module_name = name
            self.add_hooks(submodule)

    def add_hooks(self, module):
        if self.forward:
            self.fhooks.append(
                module.register_forward_hook(self.fhook_fn)
            )
        if self.backward:
            self.bhooks.append(
                module.register_full_backward_hook(self.bhook_fn)
            )

    def reset(self):
        # one report per direction is enough; the first hit is the culprit
        self.has_printed_f = self.has_printed_b = False

    def __enter__(self):
        return self

    def __exit__(self, exc_type, exc_value, exc_traceback):
        self._dump_grad_norms()
        self.close()

    def _dump_grad_norms(self):
        """Log every param grad norm when any of them went non-finite."""
        norms, bad_grads = {}, {}
        for name, param in iter(self.named_parameters):
            if param.grad is None:
                continue
            gnorm = torch.norm(param.grad.data.float(), p=2)
            norms[name] = gnorm.item()
            if not torch.isfinite(gnorm).all():
                bad_grads[name] = param.grad.data
        if bad_grads:
            logger.info("Detected nan/inf grad norm, dumping norms ...")
            logger.info(f"norms: {norms}")
            logger.info(f"gradients: {bad_grads}")

    def _describe(self, tensor, name, backward):
        """A message naming the first non-finite value, or None if clean.
        Scalar tensors (the loss) are skipped — no localization value."""
        if not torch.is_floating_point(tensor) or tensor.numel() < 2:
            return None
        with torch.no_grad():
            if bool(torch.isnan(tensor).any()):
                kind = "NaN"
            elif bool(torch.isinf(tensor).any()):
                kind = "Inf"
            else:
                return None
        direction = "backward" if backward else "forward"
        return (
            f"{kind} detected in output of {name}, "
            f"shape: {tuple(tensor.shape)}, {direction}"
        )

    def _scan(self, module, inp, value, backward):
        if torch.is_tensor(value):
            if isinstance(inp, tuple) and inp:
                inp = inp[0]
            msg = self._describe(value, module.__module_name, backward)
            if msg is None:
                return
            if not backward and torch.is_tensor(inp):
                msg += (
                    f" input max: {inp.max().item()},"
                    f" input min: {inp.min().item()}"
                )
            logger.warning(msg)
            if backward:
                self.has_printed_b = True
            else:
                self.has_printed_f = True
        elif isinstance(value, dict):
            for v in value.values():
                self._scan(module, inp, v, backward)
        elif isinstance(value, (list, tuple)):
            for v in value:
                self._scan(module, inp, v, backward)

    def fhook_fn(self, module, inp, output):
        if self.has_printed_f:
            return
        self._scan(module, inp, output, backward=False)

    def bhook_fn(self, module, inp, output):
        if self.has_printed_b:
            return
        self._scan(module, inp, output, backward=True)

    def close(self):
        for hook in [*self.fhooks, *self.bhooks]:
            hook.remove()
