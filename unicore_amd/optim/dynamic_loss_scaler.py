"""Dynamic loss scaling for fp16 training (parity: reference
unicore/optim/dynamic_loss_scaler.py:8-71).

Semantics preserved exactly: x ``scale_factor`` growth every
``scale_window`` clean updates, backoff on overflow subject to a tolerated
overflow percentage, and a FloatingPointError abort once the scale cannot
drop further (``min_loss_scale``).
"""


class DynamicLossScaler:
    def __init__(
        self,
        init_scale=2.0**15,
        scale_factor=2.0,
        scale_window=2000,
        tolerance=0.0,
        threshold=None,
        min_loss_scale=1e-4,
    ):
        self.loss_scale = init_scale
        self.scale_factor, self.scale_window = scale_factor, scale_window
        self.tolerance, self.threshold = tolerance, threshold
        self.min_loss_scale = min_loss_scale
        # update counter plus the positions of the last overflow / rescale,
        # used to rate the overflow percentage inside a window
        self._updates = 0
        self._overflow_at = -1
        self._rescale_at = -1
        self._overflow_count = 0

    def scale(self, outputs):
        """Multiply the loss (or anything) by the live scale."""
        return outputs * self.loss_scale

    def update(self) -> None:
        """Called after every applied update; grows the scale after a full
        clean window."""
        if (self._updates - self._overflow_at) % self.scale_window == 0:
            self.loss_scale = self.loss_scale * self.scale_factor
            self._rescale_at = self._updates
        self._updates += 1

    def _backoff(self) -> None:
        floor = self.threshold
        self.loss_scale = self.loss_scale / self.scale_factor
        if floor is not None:
            self.loss_scale = max(self.loss_scale, floor)

    def check_overflow(self, grad_norm) -> None:
        """Raise OverflowError (skip this update) when the grad norm is
        inf/nan; FloatingPointError when the scale has already bottomed out."""
        if grad_norm != float("inf") and grad_norm == grad_norm:
            return  # finite: nothing to do
        before = self.loss_scale
        window = self._updates - self._rescale_at

        self._overflow_at = self._updates
        self._overflow_count += 1
        if self._overflow_count / float(window) >= self.tolerance:
            self._backoff()
            self._rescale_at = self._updates
            self._overflow_count = 0

        if self.min_loss_scale >= self.loss_scale:
            # an uncommon error type the trainer can catch to stop cleanly
            self.loss_scale = before
            raise FloatingPointError(
                f"Minimum loss scale reached ({self.min_loss_scale}). Your "
                "loss is probably exploding. Try lowering the learning "
                "rate, using gradient clipping or increasing the batch size."
            )

        self._updates += 1
        raise OverflowError(f"setting loss scale to: {self.loss_scale}")
