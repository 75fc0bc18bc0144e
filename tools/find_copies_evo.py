#!/usr/bin/env python3
"""Attribute the torch elementwise/copy kernels in an Evoformer update to
their aten ops (the round-2 profile shows ~760 such launches per update)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def build_trainer():
    from unicore_amd import options, tasks
    from unicore_amd.trainer import Trainer

    argv = [
        "--task", "evoformer_synthetic", "--arch", "evoformer",
        "--loss", "masked_msa", "--optimizer", "adam",
        "--adam-betas", "(0.9, 0.99)", "--adam-eps", "1e-6",
        "--clip-norm", "0.1", "--lr-scheduler", "polynomial_decay",
        "--total-num-update", "20000", "--lr", "1e-3",
        "--batch-size", "1", "--dataset-size", "8",
        "--msa-depth", "128", "--residues", "256",
        "--bf16", "--bf16-sr", "--log-format", "none", "--num-workers", "0",
        "--seed", "7",
    ]
    parser = options.get_training_parser()
    args = options.parse_args_and_arch(parser, input_args=argv)
    args.distributed_world_size = 1
    args.distributed_rank = 0
    args.device_id = 0
    torch.manual_seed(7)
    np.random.seed(7)
    task = tasks.setup_task(args)
    task.load_dataset("train")
    model = task.build_model(args)
    loss = task.build_loss(args)
    trainer = Trainer(args, task, model, loss)
    epoch_itr = trainer.get_train_iterator(epoch=1)
    trainer.init_total_train_steps(epoch_itr)
    return trainer, epoch_itr


def main():
    trainer, epoch_itr = build_trainer()
    itr = epoch_itr.next_epoch_itr(shuffle=False)
    batches = list(itr)
    for i in range(2):
        trainer.train_step([batches[i % len(batches)]])
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        trainer.train_step([batches[0]])
        torch.cuda.synchronize()
    print("--- top 30 aten ops by CUDA time (1 update = 8 micro-steps) ---")
    for e in sorted(prof.key_averages(), key=lambda x: -x.device_time_total)[:30]:
        if e.key.startswith(("aten::", "autograd::", "Optimizer", "torch::")):
            print(f"{e.device_time_total/1000:9.2f}ms n={e.count:6d}  {e.key[:70]}")


if __name__ == "__main__":
    main()
