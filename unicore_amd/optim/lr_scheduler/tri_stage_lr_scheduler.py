"""Tri-stage (warmup/hold/decay) LR schedule (parity: reference
unicore/optim/lr_scheduler/tri_stage_lr_scheduler.py:14-177)."""

import math

from . import register_lr_scheduler
from .unicore_lr_scheduler import UnicoreLRScheduler


@register_lr_scheduler("tri_stage")
class TriStageLRSchedule(UnicoreLRScheduler):
    """Tristage learning rate schedule (SpecAugment-style):

      - warmup: linearly increase init_lr_scale*lr -> lr over warmup_steps
      - hold: keep peak lr for hold_steps
      - decay: exponential decay to final_lr_scale*lr over decay_steps
      - beyond: keep final_lr
    """

    def __init__(self, args, optimizer, total_train_steps):
        super().__init__(args, optimizer, total_train_steps)

        # calculate LR at each point
        self.peak_lr = args.lr[0]
        self.init_lr = args.init_lr_scale * args.lr[0]
        self.final_lr = args.final_lr_scale * args.lr[0]

        if args.phase_ratio is not None:
            assert args.max_update > 0
            assert sum(args.phase_ratio) == 1, "phase ratios must add up to 1"
            self.warmup_steps = int(args.max_update * args.phase_ratio[0])
            self.hold_steps = int(args.max_update * args.phase_ratio[1])
            self.decay_steps = int(args.max_update * args.phase_ratio[2])
        else:
            self.warmup_steps = args.warmup_steps
            self.hold_steps = args.hold_steps
            self.decay_steps = args.decay_steps

        assert (
            self.warmup_steps + self.hold_steps + self.decay_steps > 0
        ), "please specify steps or phase_ratio"

        self.warmup_rate = (
            (self.peak_lr - self.init_lr) / self.warmup_steps
            if self.warmup_steps != 0
            else 0
        )
        self.decay_factor = -math.log(args.final_lr_scale) / self.decay_steps

        # initial learning rate
        self.lr = self.init_lr
        self.optimizer.set_lr(self.lr)

    @classmethod
    def add_args(cls, parser):
        """Add arguments to the parser for this LR scheduler."""
        parser.add_argument(
            "--warmup-steps",
            default=4000,
            type=int,
            metavar="N",
            help="warmup the learning rate linearly for the first N updates",
        )
        parser.add_argument(
            "--hold-steps",
            default=20000,
            type=int,
            metavar="N",
            help="steps in hold stage",
        )
        parser.add_argument(
            "--decay-steps",
            default=60000,
            type=int,
            metavar="N",
            help="steps in decay stages",
        )
        parser.add_argument(
            "--init-lr-scale",
            default=0.01,
            type=float,
            help="initial learning rate scale during warmup phase",
        )
        parser.add_argument(
            "--final-lr-scale",
            default=0.01,
            type=float,
            help="final learning rate scale",
        )
        parser.add_argument(
            "--phase-ratio",
            default=None,
            type=lambda x: eval(x),
            help="automatically sets warmup/hold/decay steps to the ratio "
            "specified here from max_updates (requires --max-update)",
        )

    def _decide_stage(self, update_step):
        """Return (stage index, corresponding offset into the stage)."""
        if update_step < self.warmup_steps:
            return 0, update_step

        offset = self.warmup_steps
        if update_step < offset + self.hold_steps:
            return 1, update_step - offset

        offset += self.hold_steps
        if update_step <= offset + self.decay_steps:
            # decay stage
            return 2, update_step - offset

        offset += self.decay_steps
        # still here ? constant lr stage
        return 3, update_step - offset

    def step(self, epoch, val_loss=None):
        """Update the learning rate at the end of the given epoch."""
        super().step(epoch, val_loss)
        # we don't change the learning rate at epoch boundaries
        return self.optimizer.get_lr()

    def step_update(self, num_updates):
        """Update the learning rate after each update."""
        stage, steps_in_stage = self._decide_stage(num_updates)
        if stage == 0:
            self.lr = self.init_lr + self.warmup_rate * steps_in_stage
        elif stage == 1:
            self.lr = self.peak_lr
        elif stage == 2:
            self.lr = self.peak_lr * math.exp(-self.decay_factor * steps_in_stage)
        elif stage == 3:
            self.lr = self.final_lr
        else:
            raise ValueError("Undefined stage")

        self.optimizer.set_lr(self.lr)
        return self.lr
