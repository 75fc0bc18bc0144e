#!/usr/bin/env python3
"""Attribute the per-layer direct_copy / copyBuffer kernels in the BERT
bench step to their Python call sites (torch.profiler with stacks)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def build_trainer():
    from unicore_amd import options, tasks
    from unicore_amd.trainer import Trainer

    argv = [
        "--task", "bert_synthetic", "--arch", "bert_base",
        "--loss", "masked_lm", "--optimizer", "adam",
        "--adam-betas", "(0.9, 0.98)", "--adam-eps", "1e-6",
        "--clip-norm", "1.0", "--lr-scheduler", "polynomial_decay",
        "--total-num-update", "1000", "--lr", "1e-4",
        "--batch-size", "127", "--dataset-size", "508",
        "--tokens-per-sample", "512", "--max-seq-len", "514",
        "--bf16", "--log-format", "none", "--num-workers", "0",
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
    for i in range(3):
        trainer.train_step([batches[i % len(batches)]])
    torch.cuda.synchronize()
    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=True) as prof:
        for i in range(2):
            trainer.train_step([batches[i % len(batches)]])
        torch.cuda.synchronize()
    evs = prof.key_averages(group_by_input_shape=True)
    rows = [e for e in evs
            if ("copy" in e.key.lower() or "contiguous" in e.key
                or "clone" in e.key or "Memcpy" in e.key)
            and e.device_time_total > 0]
    rows.sort(key=lambda e: -e.device_time_total)
    for e in rows[:14]:
        print(f"== {e.key[:70]}  n={e.count}  "
              f"cuda_total={e.device_time_total/1000:.2f}ms  "
              f"shapes={e.input_shapes}")
    # and the aggregate top ops for orientation
    print("\n--- top 20 ops by CUDA time ---")
    for e in sorted(prof.key_averages(), key=lambda x: -x.device_time_total)[:20]:
        print(f"{e.device_time_total/1000:9.2f}ms n={e.count:5d}  {e.key[:80]}")


if __name__ == "__main__":
    main()
