"""Masked-LM loss (parity: reference unicore/losses/masked_lm.py:12-67).

Targets are pad everywhere except masked positions; the model is given the
mask so its LM head only projects the masked rows, and the loss runs
through the fused online-logsumexp cross entropy.
"""

import math

import torch

from unicore_amd import metrics
from unicore_amd.losses import UnicoreLoss, register_loss
from unicore_amd.modules.cross_entropy import fused_nll_loss


@register_loss("masked_lm")
class MaskedLMLoss(UnicoreLoss):
    def __init__(self, task):
        super().__init__(task)
        self.padding_idx = task.dictionary.pad()

    def forward(self, model, sample, reduce=True):
        target = sample["target"]
        masked = target.ne(self.padding_idx)
        sample_size = masked.int().sum()
        # degenerate batches with zero masked tokens still need one row so
        # the graph stays connected (its target is pad -> zero loss)
        masked = torch.where(masked.any(), masked, masked.new([True]))

        logits = model(**sample["net_input"], masked_tokens=masked)
        loss = fused_nll_loss(
            logits, target[masked], ignore_index=self.padding_idx
        )
        stats = {
            "loss": loss.data,
            "bsz": target.size(0),
            "sample_size": sample_size,
            "seq_len": target.size(1) * target.size(0),
        }
        return loss, sample_size, stats

    @staticmethod
    def reduce_metrics(logging_outputs, split="valid") -> None:
        totals = {
            key: sum(log.get(key, 0) for log in logging_outputs)
            for key in ("loss", "bsz", "sample_size", "seq_len")
        }
        n = totals["sample_size"]
        denom = n if n > 0 else 1  # a whole batch can have zero masked tokens
        # log(2): report in bits rather than nats
        metrics.log_scalar("loss", totals["loss"] / denom / math.log(2), n,
                           round=3)
        metrics.log_scalar("seq_len", totals["seq_len"] / totals["bsz"], 1,
                           round=3)

    @staticmethod
    def logging_outputs_can_be_summed(is_train) -> bool:
        return True
