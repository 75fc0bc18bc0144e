"""Attribute eager elementwise kernels in the BERT bench to source lines
(torch.profiler with stacks over a few training steps)."""
import logging
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main():
    logging.basicConfig(level=logging.WARNING)
    import bench

    class B:
        batch_size = 48
        seq_len = 512
        model = "bert_base"
        cpu = False
        eager = False
        ddp_backend = "c10d"

    args = bench.build_framework_args(B, 1, 0, 0)
    torch.cuda.set_device(0)
    torch.manual_seed(args.seed)
    np.random.seed(args.seed)

    from unicore_amd import tasks
    from unicore_amd.trainer import Trainer

    task = tasks.setup_task(args)
    task.load_dataset("train")
    model = task.build_model(args)
    loss = task.build_loss(args)
    trainer = Trainer(args, task, model, loss)
    epoch_itr = trainer.get_train_iterator(epoch=1)
    trainer.init_total_train_steps(epoch_itr)
    itr = epoch_itr.next_epoch_itr(shuffle=False)
    batches = []
    for i, sample in enumerate(itr):
        batches.append(sample)
        if i >= 3:
            break

    for i in range(2):
        trainer.train_step([batches[i % len(batches)]])
    torch.cuda.synchronize()

    from torch.profiler import ProfilerActivity, profile

    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
                 with_stack=True) as prof:
        for i in range(2):
            trainer.train_step([batches[i % len(batches)]])
        torch.cuda.synchronize()

    ka = prof.key_averages(group_by_stack_n=7)
    rows = [e for e in ka if e.device_time_total > 0]
    rows.sort(key=lambda e: -e.device_time_total)
    shown = 0
    for e in rows:
        k = e.key
        if not any(s in k for s in ("add", "copy", "sum", "mul", "div", "cat")):
            continue
        print(f"{e.device_time_total/1e3:9.2f}ms {e.count:5d}x  {k[:80]}")
        for fr in (e.stack or [])[:7]:
            print(f"      {fr}")
        shown += 1
        if shown >= 12:
            break


if __name__ == "__main__":
    main()
