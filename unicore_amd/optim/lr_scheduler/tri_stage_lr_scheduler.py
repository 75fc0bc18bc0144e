"""Tri-stage (warmup / hold / exponential-decay) LR schedule (parity:
reference unicore/optim/lr_scheduler/tri_stage_lr_scheduler.py:14-177;
SpecAugment-style).

Stages: ramp ``init_lr_scale*lr -> lr`` over warmup_steps, hold the peak
for hold_steps, decay exponentially to ``final_lr_scale*lr`` over
decay_steps, then hold the final LR. ``--phase-ratio`` derives the three
stage lengths from ``--max-update`` instead.
"""

import math

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("tri_stage")
class TriStageLRSchedule(UnicoreLRScheduler):
    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)

        self.peak_lr = args.lr[0]
        self.init_lr = args.init_lr_scale * args.lr[0]
        self.final_lr = args.final_lr_scale * args.lr[0]

        if args.phase_ratio is not None:
            assert args.max_update > 0
            assert sum(args.phase_ratio) == 1, "phase ratios must add up to 1"
            lengths = [int(args.max_update * r) for r in args.phase_ratio]
            self.warmup_steps, self.hold_steps, self.decay_steps = lengths
        else:
            self.warmup_steps = args.warmup_steps
            self.hold_steps = args.hold_steps
            self.decay_steps = args.decay_steps

        assert self.warmup_steps + self.hold_steps + self.decay_steps > 0, \
            "please specify steps or phase_ratio"

        self.warmup_rate = (
            (self.peak_lr - self.init_lr) / self.warmup_steps
            if self.warmup_steps != 0
            else 0
        )
        # decay constant chosen so the stage ends exactly at final_lr
        self.decay_factor = -math.log(args.final_lr_scale) / self.decay_steps

        self.lr = self.init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        parser.add_argument("--warmup-steps", default=4000, type=int,
                            metavar="N", help="length of the linear ramp")
        parser.add_argument("--hold-steps", default=20000, type=int,
                            metavar="N", help="length of the peak plateau")
        parser.add_argument("--decay-steps", default=60000, type=int,
                            metavar="N", help="length of the decay stage")
        parser.add_argument("--init-lr-scale", default=0.01, type=float,
                            help="starting LR as a fraction of peak")
        parser.add_argument("--final-lr-scale", default=0.01, type=float,
                            help="ending LR as a fraction of peak")
        parser.add_argument("--phase-ratio", default=None,
                            type=lambda x: eval(x),
                            help="derive warmup/hold/decay lengths from "
                                 "--max-update by these ratios")

    def _decide_stage(self, update_step):
        """(stage index, offset within that stage) for an update count."""
        boundary = self.warmup_steps
        if update_step < boundary:
            return 0, update_step
        if update_step < boundary + self.hold_steps:
            return 1, update_step - boundary
        boundary += self.hold_steps
        if update_step <= boundary + self.decay_steps:
            return 2, update_step - boundary
        boundary += self.decay_steps
        return 3, update_step - boundary  # final constant stage

    def step(self, epoch, val_loss=None):
        super().step(epoch, val_loss)
        # update-driven schedule: epoch boundaries change nothing
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        stage, offset = self._decide_stage(num_updates)
        if stage == 0:
            self.lr = self.init_lr + self.warmup_rate * offset
        elif stage == 1:
            self.lr = self.peak_lr
        elif stage == 2:
            self.lr = self.peak_lr * math.exp(-self.decay_factor * offset)
        elif stage == 3:
            self.lr = self.final_lr
        else:
            raise ValueError("Undefined stage")
        self.optimizer.set_lr(self.lr)
        return self.lr
