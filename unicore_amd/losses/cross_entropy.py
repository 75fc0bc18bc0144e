"""Plain cross entropy loss (parity: reference
unicore/losses/cross_entropy.py:13-65). Sample size = batch rows."""

import math

import torch
import torch.nn.functional as F

from unicore_amd import metrics
from unicore_amd.losses import UnicoreLoss, register_loss


@register_loss("cross_entropy")
class CrossEntropyLoss(UnicoreLoss):
    def __init__(self, task):
        super().__init__(task)

    def forward(self, model, sample, reduce=True):
        net_output = model(**sample["net_input"])
        loss = self.compute_loss(model, net_output, sample, reduce=reduce)
        sample_size = sample["target"].size(0)
        stats = {
            "loss": loss.data,
            "bsz": sample_size,
            "sample_size": sample_size,
        }
        return loss, sample_size, stats

    def compute_loss(self, model, net_output, sample, reduce=True):
        # fp32 log-softmax for numerical stability in low-precision training
        logp = F.log_softmax(net_output, dim=-1, dtype=torch.float32)
        return F.nll_loss(
            logp.view(-1, logp.size(-1)),
            sample["target"].view(-1),
            reduction="sum" if reduce else "none",
        )

    @staticmethod
    def reduce_metrics(logging_outputs, split="valid") -> None:
        loss_total = sum(log.get("loss", 0) for log in logging_outputs)
        n = sum(log.get("sample_size", 0) for log in logging_outputs)
        # log(2): report in bits rather than nats
        metrics.log_scalar("loss", loss_total / n / math.log(2), n, round=3)

    @staticmethod
    def logging_outputs_can_be_summed(is_train) -> bool:
        return True
