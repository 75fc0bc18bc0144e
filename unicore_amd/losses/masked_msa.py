"""Masked-MSA cross entropy for the Evoformer stress config — masked_lm
semantics over (B, S, L) token grids."""

import math

import torch

from unicore_amd import metrics
from unicore_amd.losses import UnicoreLoss, register_loss
from unicore_amd.modules.cross_entropy import fused_nll_loss


@register_loss("masked_msa")
class MaskedMSALoss(UnicoreLoss):
    def __init__(self, task):
        super().__init__(task)
        self.padding_idx = task.dictionary.pad()

    def forward(self, model, sample, reduce=True):
        logits = model(**sample["net_input"])  # (B, S, L, V)
        target = sample["target"]
        masked = target.ne(self.padding_idx)
        sample_size = masked.int().sum()
        masked = torch.where(masked.any(), masked, masked.new([True]))
        loss = fused_nll_loss(logits[masked], target[masked], ignore_index=self.padding_idx)
        logging_output = {
            "loss": loss.data,
            "bsz": target.size(0),
            "sample_size": sample_size,
        }
        return loss, sample_size, logging_output

    @staticmethod
    def reduce_metrics(logging_outputs, split="valid") -> None:
        loss_sum = sum(log.get("loss", 0) for log in logging_outputs)
        sample_size = sum(log.get("sample_size", 0) for log in logging_outputs)
        # sample_size can be 0 when no tokens were masked in the whole batch
        denom = sample_size if sample_size > 0 else 1
        metrics.log_scalar(
            "loss", loss_sum / denom / math.log(2), sample_size, round=3
        )

    @staticmethod
    def logging_outputs_can_be_summed(is_train) -> bool:
        return True
