"""Stress-config model tests (BASELINE.json configs 4 & 5): the pair-bias
3D molecular transformer (mol_pairbias + unimol_synthetic + mol_pretrain)
and the Evoformer block (evoformer + evoformer_synthetic + masked_msa).
CPU smoke here; GPU variants exercise the fused bias-softmax / RMSNorm /
bf16-SR paths.
"""

import sys

import numpy as np
import pytest
import torch

from unicore_cli import train as train_cli

MOL_ARGV = [
    "--task", "unimol_synthetic",
    "--arch", "mol_pairbias",
    "--loss", "mol_pretrain",
    "--optimizer", "adam",
    "--lr-scheduler", "fixed",
    "--lr", "1e-4",
    "--batch-size", "4",
    "--dataset-size", "16",
    "--atoms-per-mol", "32",
    "--encoder-layers", "2",
    "--encoder-embed-dim", "64",
    "--encoder-ffn-embed-dim", "128",
    "--encoder-attention-heads", "2",
    "--gaussian-kernels", "16",
    "--log-format", "none",
    "--num-workers", "0",
    "--no-save",
]

EVO_ARGV = [
    "--task", "evoformer_synthetic",
    "--arch", "evoformer",
    "--loss", "masked_msa",
    "--optimizer", "adam",
    "--lr-scheduler", "fixed",
    "--lr", "1e-4",
    "--batch-size", "2",
    "--dataset-size", "8",
    "--msa-depth", "8",
    "--residues", "24",
    "--evo-layers", "2",
    "--msa-dim", "64",
    "--pair-dim", "32",
    "--evo-heads", "4",
    "--log-format", "none",
    "--num-workers", "0",
    "--no-save",
]


def _run(argv, monkeypatch, tmp_path):
    monkeypatch.setattr(
        sys, "argv",
        ["unicore-train"] + argv + ["--save-dir", str(tmp_path / "sv")],
    )
    train_cli.cli_main()


def test_mol_pairbias_cpu(tmp_path, monkeypatch):
    _run(MOL_ARGV + ["--max-update", "3", "--max-epoch", "1", "--cpu"],
         monkeypatch, tmp_path)


def test_evoformer_cpu(tmp_path, monkeypatch):
    _run(EVO_ARGV + ["--max-update", "3", "--max-epoch", "1", "--cpu"],
         monkeypatch, tmp_path)


def test_mol_model_shapes_cpu():
    from unicore_amd import options, tasks

    parser = options.get_training_parser()
    args = options.parse_args_and_arch(
        parser, input_args=MOL_ARGV + ["--cpu", "--save-dir", "/tmp/x"]
    )
    task = tasks.setup_task(args)
    model = task.build_model(args)
    B, L = 2, 16
    toks = torch.randint(5, 20, (B, L))
    coords = torch.randn(B, L, 3)
    logits, delta = model(toks, coords)
    assert logits.shape == (B, L, len(task.dictionary))
    assert delta.shape == (B, L, 3)
    loss = logits.float().pow(2).mean() + delta.pow(2).mean()
    loss.backward()


def test_evoformer_shapes_cpu():
    from unicore_amd import options, tasks

    parser = options.get_training_parser()
    args = options.parse_args_and_arch(
        parser, input_args=EVO_ARGV + ["--cpu", "--save-dir", "/tmp/x"]
    )
    task = tasks.setup_task(args)
    model = task.build_model(args)
    B, S, L = 2, 4, 12
    toks = torch.randint(5, 20, (B, S, L))
    logits = model(toks)
    assert logits.shape == (B, S, L, len(task.dictionary))
    logits.float().pow(2).mean().backward()


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_mol_pairbias_gpu_bf16(tmp_path, monkeypatch):
    _run(
        MOL_ARGV
        + ["--max-update", "4", "--max-epoch", "1", "--bf16",
           "--atoms-per-mol", "64", "--encoder-embed-dim", "128",
           "--encoder-attention-heads", "4"],
        monkeypatch, tmp_path,
    )


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_evoformer_gpu_bf16_sr_grad_accum(tmp_path, monkeypatch):
    """Config 5: bf16 + stochastic rounding + grad-accum 8."""
    _run(
        EVO_ARGV
        + ["--max-update", "3", "--bf16", "--bf16-sr", "--update-freq", "8"],
        monkeypatch, tmp_path,
    )


@pytest.mark.gpu
def test_evoformer_hip_graph_blocks_parity():
    """hipGraph-captured Evoformer blocks must produce the same loss
    trajectory as the eager block stack (dropout 0, static shapes)."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs an MI355X")
    from unicore_amd import options, tasks
    from unicore_amd.trainer import Trainer

    def run(graphs):
        argv = [
            "--task", "evoformer_synthetic",
            "--arch", "evoformer",
            "--loss", "masked_msa",
            "--optimizer", "adam",
            "--lr-scheduler", "fixed",
            "--lr", "1e-3",
            "--batch-size", "1",
            "--dataset-size", "8",
            "--msa-depth", "16",
            "--residues", "48",
            "--evo-layers", "2",
            "--dropout", "0.0",
            "--seed", "5",
            "--bf16",
            "--log-format", "none",
            "--num-workers", "0",
        ] + (["--hip-graph-blocks"] if graphs else [])
        parser = options.get_training_parser()
        args = options.parse_args_and_arch(parser, input_args=argv)
        args.distributed_world_size = 1
        args.distributed_rank = 0
        args.device_id = 0
        args.distributed_no_spawn = True
        torch.manual_seed(args.seed)
        task = tasks.setup_task(args)
        task.load_dataset("train")
        model = task.build_model(args)
        loss = task.build_loss(args)
        trainer = Trainer(args, task, model, loss)
        epoch_itr = trainer.get_train_iterator(epoch=1)
        trainer.init_total_train_steps(epoch_itr)
        itr = epoch_itr.next_epoch_itr(shuffle=False)
        out = []
        for _ in range(3):
            log = trainer.train_step([next(itr)])
            out.append(float(log["loss"]))
        return out

    plain = run(False)
    graphed = run(True)
    assert graphed == pytest.approx(plain, rel=2e-2), (plain, graphed)


@pytest.mark.gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_evoformer_fused_preln_chain_parity(monkeypatch):
    """The EvoformerBlock's fused pre-LN residual chain (dropout_add_ln_pre
    joins emitting summed + normed) must match the eager composition of the
    same math (UNICORE_FUSED_LN_JOIN=0 falls back inside the wrapper) for
    outputs AND parameter gradients."""
    from unicore_amd.models.evoformer import EvoformerBlock

    def run(fused):
        monkeypatch.setenv("UNICORE_FUSED_LN_JOIN", "1" if fused else "0")
        torch.manual_seed(5)
        blk = EvoformerBlock(d_msa=64, d_pair=32, heads=4, dropout=0.0)
        blk = blk.cuda().bfloat16().train()
        torch.manual_seed(9)
        msa = torch.randn(1, 8, 16, 64, device="cuda", dtype=torch.bfloat16)
        pair = torch.randn(1, 16, 16, 32, device="cuda", dtype=torch.bfloat16)
        m, p = blk(msa, pair)
        (m.float().square().mean() + p.float().square().mean()).backward()
        grads = {k: v.grad.float().clone() for k, v in
                 blk.named_parameters() if v.grad is not None}
        return m.float(), p.float(), grads

    m1, p1, g1 = run(True)
    m0, p0, g0 = run(False)
    assert torch.allclose(m1, m0, atol=3e-2, rtol=3e-2), (m1 - m0).abs().max()
    assert torch.allclose(p1, p0, atol=3e-2, rtol=3e-2), (p1 - p0).abs().max()
    assert set(g1) == set(g0)
    for k in g1:
        assert torch.allclose(g1[k], g0[k], atol=5e-2, rtol=5e-2), \
            (k, (g1[k] - g0[k]).abs().max())
