"""Masked-LM loss (parity: reference unicore/losses/masked_lm.py:12-67).

Targets are pad everywhere except masked positions; the model is given the
mask so its LM head only projects the masked rows, and the loss runs
through the fused online-logsumexp cross entropy.

Pad-to-bucket (GPU default, UNICORE_LMHEAD_BUCKET, 0 disables): the number
of selected rows is rounded UP to a bucket multiple with filler rows whose
target is pad (ignored by the CE). The lm-head GEMM shapes then stop
varying with the per-batch mask count (9748/9751/... rows), which makes
them offline-tunable and keeps hipBLASLt on one algorithm.
"""

import math
import os

import torch

from unicore_amd import metrics
from unicore_amd.losses import UnicoreLoss, register_loss
from unicore_amd.modules.cross_entropy import fused_nll_loss


@register_loss("masked_lm")
class MaskedLMLoss(UnicoreLoss):
    def __init__(self, task):
        super().__init__(task)
        self.padding_idx = task.dictionary.pad()

    def _bucketed_selection(self, target, masked):
        """(flat indices padded to a bucket multiple, matching targets).
        Filler rows duplicate index 0 but carry a pad target, so the CE
        ignores them; the GEMM row count becomes batch-invariant."""
        bucket = int(os.environ.get("UNICORE_LMHEAD_BUCKET", "512"))
        if bucket <= 1 or not target.is_cuda:
            return None, None
        flat_idx = masked.view(-1).nonzero(as_tuple=False).squeeze(1)
        n = flat_idx.numel()
        if n == 0:
            return None, None
        want = min(-(-n // bucket) * bucket, target.numel())
        picked_targets = target.view(-1).index_select(0, flat_idx)
        if want > n:
            filler = flat_idx.new_zeros(want - n)
            flat_idx = torch.cat([flat_idx, filler])
            picked_targets = torch.cat([
                picked_targets,
                picked_targets.new_full((want - n,), self.padding_idx),
            ])
        return flat_idx, picked_targets

    def forward(self, model, sample, reduce=True):
        target = sample["target"]
        masked = target.ne(self.padding_idx)
        sample_size = masked.int().sum()

        sel_idx, sel_tgt = self._bucketed_selection(target, masked)
        if sel_idx is not None:
            logits = model(**sample["net_input"], masked_tokens=sel_idx)
            loss = fused_nll_loss(
                logits, sel_tgt, ignore_index=self.padding_idx
            )
        else:
            # degenerate batches with zero masked tokens still need one row
            # so the graph stays connected (its target is pad -> zero loss)
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
