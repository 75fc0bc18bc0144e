"""Masked LM loss (parity: reference unicore/losses/masked_lm.py:12-67)."""

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
        """Compute the loss for the given sample.

        Returns a tuple with three elements:
        1) the loss
        2) the sample size, which is used as the denominator for the gradient
        3) logging outputs to display while training
        """
        masked_tokens = sample["target"].ne(self.padding_idx)
        sample_size = masked_tokens.int().sum()

        masked_tokens = torch.where(
            masked_tokens.any(),
            masked_tokens,
            masked_tokens.new([True]),
        )
        logits = model(**sample["net_input"], masked_tokens=masked_tokens)
        target = sample["target"]
        if masked_tokens is not None:
            target = target[masked_tokens]
        loss = fused_nll_loss(logits, target, ignore_index=self.padding_idx)
        logging_output = {
            "loss": loss.data,
            "bsz": sample["target"].size(0),
            "sample_size": sample_size,
            "seq_len": sample["target"].size(1) * sample["target"].size(0),
        }
        return loss, sample_size, logging_output

    @staticmethod
    def reduce_metrics(logging_outputs, split="valid") -> None:
        """Aggregate logging outputs from data parallel training."""
        loss_sum = sum(log.get("loss", 0) for log in logging_outputs)
        bsz = sum(log.get("bsz", 0) for log in logging_outputs)
        sample_size = sum(log.get("sample_size", 0) for log in logging_outputs)
        seq_len = sum(log.get("seq_len", 0) for log in logging_outputs)
        # we divide by log(2) to convert the loss from base e to base 2
        metrics.log_scalar(
            "loss", loss_sum / sample_size / math.log(2), sample_size, round=3
        )
        metrics.log_scalar("seq_len", seq_len / bsz, 1, round=3)

    @staticmethod
    def logging_outputs_can_be_summed(is_train) -> bool:
        """
        Whether the logging outputs returned by `forward` can be summed
        across workers prior to calling `reduce_metrics`. Setting this
        to True will improve distributed training speed.
        """
        return True
