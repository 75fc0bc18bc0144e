"""Masked-atom + coordinate-denoising loss for the mol_pairbias model
(BASELINE.json stress config 4).  Demonstrates a composite multi-task
UnicoreLoss: token cross-entropy on masked atoms + L2 on predicted
coordinate deltas (the Uni-Mol pretraining recipe shape)."""

import math

import torch

from unicore_amd import metrics
from unicore_amd.losses import UnicoreLoss, register_loss
from unicore_amd.modules.cross_entropy import fused_nll_loss


@register_loss("mol_pretrain")
class MolPretrainLoss(UnicoreLoss):
    def __init__(self, task):
        super().__init__(task)
        self.padding_idx = task.dictionary.pad()
        self.coord_loss_weight = getattr(task.args, "coord_loss_weight", 1.0)

    def forward(self, model, sample, reduce=True):
        logits, coord_delta = model(**sample["net_input"])
        target = sample["target"]
        masked = target.ne(self.padding_idx)
        sample_size = masked.int().sum()
        masked = torch.where(masked.any(), masked, masked.new([True]))
        token_loss = fused_nll_loss(logits[masked], target[masked], ignore_index=self.padding_idx)
        # coordinate denoising: predict the delta back to clean positions
        coord_target = sample["coord_target"].float()
        atom_mask = sample["net_input"]["src_tokens"].ne(self.padding_idx)
        diff = (coord_delta - coord_target) * atom_mask.unsqueeze(-1)
        coord_loss = diff.pow(2).sum() / 3.0
        loss = token_loss + self.coord_loss_weight * coord_loss
        logging_output = {
            "loss": loss.data,
            "token_loss": token_loss.data,
            "coord_loss": coord_loss.data,
            "bsz": target.size(0),
            "sample_size": sample_size,
            "seq_len": target.size(1) * target.size(0),
        }
        return loss, sample_size, logging_output

    @staticmethod
    def reduce_metrics(logging_outputs, split="valid") -> None:
        loss_sum = sum(log.get("loss", 0) for log in logging_outputs)
        token_sum = sum(log.get("token_loss", 0) for log in logging_outputs)
        coord_sum = sum(log.get("coord_loss", 0) for log in logging_outputs)
        sample_size = sum(log.get("sample_size", 0) for log in logging_outputs)
        # sample_size can be 0 when no tokens were masked in the whole batch
        denom = sample_size if sample_size > 0 else 1
        metrics.log_scalar(
            "loss", loss_sum / denom / math.log(2), sample_size, round=3
        )
        metrics.log_scalar(
            "token_loss", token_sum / denom / math.log(2), sample_size, round=3
        )
        metrics.log_scalar("coord_loss", coord_sum / denom, sample_size, round=3)

    @staticmethod
    def logging_outputs_can_be_summed(is_train) -> bool:
        return True
