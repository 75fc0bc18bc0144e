"""Checkpoint format compatibility with the reference on-disk schema.

Two directions (reference unicore/trainer.py:258-284):
1. a committed fixture written field-by-field in the reference layout loads
   through our trainer (weights, update count, iterator epoch, best metric);
2. a checkpoint our trainer writes carries exactly the reference's key
   structure, so the reference could read it back.
"""

import os

import pytest
import torch


FIXTURE = os.path.join(
    os.path.dirname(__file__), "fixtures", "reference_format_checkpoint.pt"
)

TINY_ARGV = [
    "--task", "bert_synthetic",
    "--arch", "bert_base",
    "--loss", "masked_lm",
    "--optimizer", "adam",
    "--lr-scheduler", "fixed",
    "--lr", "1e-4",
    "--batch-size", "2",
    "--dataset-size", "8",
    "--tokens-per-sample", "64",
    "--max-seq-len", "66",
    "--vocab-size", "128",
    "--encoder-layers", "2",
    "--encoder-embed-dim", "64",
    "--encoder-ffn-embed-dim", "128",
    "--encoder-attention-heads", "2",
    "--log-format", "none",
    "--num-workers", "0",
    "--seed", "11",
    "--cpu",
]


def _build_trainer(tmp_path, extra=()):
    from unicore_amd import options, tasks
    from unicore_amd.trainer import Trainer

    argv = TINY_ARGV + ["--save-dir", str(tmp_path)] + list(extra)
    parser = options.get_training_parser()
    args = options.parse_args_and_arch(parser, input_args=argv)
    torch.manual_seed(args.seed)
    task = tasks.setup_task(args)
    task.load_dataset("train")
    model = task.build_model(args)
    loss = task.build_loss(args)
    return args, task, Trainer(args, task, model, loss)


def test_reference_format_fixture_loads(tmp_path):
    """The committed reference-layout checkpoint restores through
    trainer.load_checkpoint."""
    assert os.path.isfile(FIXTURE), (
        "fixture missing; regenerate with tools/make_ckpt_fixture.py"
    )
    args, task, trainer = _build_trainer(tmp_path)

    extra_state, epoch_itr = trainer.load_checkpoint(FIXTURE)

    # weights from the fixture landed in the model
    want = torch.load(FIXTURE, map_location="cpu", weights_only=False)
    got = trainer.get_model().state_dict()
    for key, tensor in want["model"].items():
        assert torch.equal(got[key], tensor), key

    # training position and best metric round-tripped
    assert extra_state["best"] == pytest.approx(2.5)
    assert extra_state["train_iterator"]["epoch"] == 2
    assert epoch_itr.epoch == 2
    # optimizer state was absent -> num_updates untouched (0), but the
    # optimizer history is available for the next save
    assert trainer._optim_history[-1]["num_updates"] == 7


def test_saved_schema_matches_reference(tmp_path):
    """One real save must produce exactly the reference's key structure."""
    args, task, trainer = _build_trainer(tmp_path)
    epoch_itr = trainer.get_train_iterator(epoch=1)
    trainer.init_total_train_steps(epoch_itr)
    itr = epoch_itr.next_epoch_itr(shuffle=False)
    trainer.train_step([next(itr)])

    path = str(tmp_path / "schema_probe.pt")
    trainer.save_checkpoint(
        path, {"train_iterator": epoch_itr.state_dict(), "val_loss": 1.0}
    )
    state = torch.load(path, map_location="cpu", weights_only=False)

    # top-level keys (reference unicore/trainer.py:258-284)
    assert set(state.keys()) == {
        "args", "model", "loss", "optimizer_history", "task_state",
        "extra_state", "last_optimizer_state",
    }
    history = state["optimizer_history"][-1]
    assert set(history.keys()) == {
        "loss_name", "optimizer_name", "lr_scheduler_state", "num_updates"
    }
    assert history["loss_name"] == "MaskedLMLoss"
    for key in ("metrics", "previous_training_time", "train_iterator",
                "val_loss"):
        assert key in state["extra_state"], key
    it = state["extra_state"]["train_iterator"]
    assert {"epoch", "iterations_in_epoch", "shuffle", "len"} <= set(it.keys())
    # model keys carry no wrapper prefixes
    assert all(not k.startswith(("module.", "_wrapped"))
               for k in state["model"])
