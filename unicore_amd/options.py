"""CLI argument system: two-pass parsing with plugin-injected args.

Parity: reference unicore/options.py — get_training_parser:24,
parse_args_and_arch:43, get_parser:159, add_dataset_args:229,
add_distributed_training_args:269, add_optimization_args:314,
add_checkpoint_args:342. Flag names and defaults follow SURVEY.md
Appendix C; the flag groups are declared as data tables here and installed
by one loop per group.
"""

import argparse
from typing import Callable, List, Optional

import torch

from unicore_amd import utils


def eval_str_list(x, type=float):
    """Parse '0.1' / '0.1,0.2' / python-literal lists into a typed list."""
    if x is None:
        return None
    if isinstance(x, str):
        try:
            x = eval(x)
        except Exception:
            x = [type(v) for v in x.split(",")]
    try:
        return list(map(type, x))
    except TypeError:
        return [type(x)]


# kept under its old name for callers/tests that import it
utils_eval_str_list = eval_str_list


def _install(target, table):
    for flags, spec in table:
        if isinstance(flags, str):
            flags = (flags,)
        target.add_argument(*flags, **spec)


# ---------------------------------------------------------------------------
# flag tables
# ---------------------------------------------------------------------------

_COMMON_FLAGS = [
    ("--no-progress-bar", dict(action="store_true",
                               help="disable progress bar")),
    ("--log-interval", dict(type=int, default=1000, metavar="N",
                            help="log every N batches when the bar is off")),
    ("--log-format", dict(default=None,
                          choices=["json", "none", "simple", "tqdm"],
                          help="log format to use")),
    ("--tensorboard-logdir", dict(metavar="DIR", default="",
                                  help="tensorboard log destination")),
    ("--wandb-project", dict(metavar="WANDB", default="",
                             help="wandb project name (empty disables wandb)")),
    ("--seed", dict(default=1, type=int, metavar="N",
                    help="RNG seed")),
    ("--cpu", dict(action="store_true", help="run on CPU instead of GPU")),
    ("--fp16", dict(action="store_true", help="train in FP16")),
    ("--bf16", dict(action="store_true", help="train in BF16")),
    ("--bf16-sr", dict(action="store_true",
                       help="stochastic rounding on the fp32->bf16 master "
                            "weight writeback")),
    ("--allreduce-fp32-grad", dict(action="store_true",
                                   help="all-reduce the fp32 master grads "
                                        "instead of the low-precision grads "
                                        "(requires --ddp-backend no_c10d)")),
    ("--fp16-no-flatten-grads", dict(action="store_true",
                                     help="(unsupported escape hatch kept "
                                          "for CLI parity)")),
    ("--memory-efficient-fp16", dict(action="store_true",
                                     help="(kept for CLI parity)")),
    ("--fp16-init-scale", dict(default=2**7, type=int,
                               help="starting FP16 loss scale")),
    ("--fp16-scale-window", dict(type=int,
                                 help="clean updates between scale raises")),
    ("--fp16-scale-tolerance", dict(default=0.0, type=float,
                                    help="tolerated overflow fraction before "
                                         "lowering the loss scale")),
    ("--min-loss-scale", dict(default=1e-4, type=float, metavar="D",
                              help="abort training once the FP16 loss scale "
                                   "falls to this value")),
    ("--threshold-loss-scale", dict(type=float,
                                    help="lower bound on the FP16 loss scale")),
    ("--user-dir", dict(default=None,
                        help="python module directory with custom "
                             "extensions (tasks / architectures)")),
    ("--empty-cache-freq", dict(default=0, type=int,
                                help="clear the CUDA cache every N steps "
                                     "(0 disables)")),
    ("--all-gather-list-size", dict(default=16384, type=int,
                                    help="byte budget per rank for pickled "
                                         "stat gathering")),
    ("--suppress-crashes", dict(action="store_true",
                                help="catch crashes in the entry point so "
                                     "sweeps can read a return value")),
    ("--profile", dict(action="store_true",
                       help="emit autograd profiler ranges (rocTX under "
                            "rocprof)")),
    ("--gemm-tuning-file", dict(default=None, metavar="CSV",
                                help="offline TunableOp results to load "
                                     "(see tools/tunableop_pershape.sh)")),
    ("--ema-decay", dict(default=-1.0, type=float,
                         help="EMA decay for model params (<0 disables)")),
    ("--validate-with-ema", dict(action="store_true")),
]

_DATASET_FLAGS = [
    ("--num-workers", dict(default=1, type=int, metavar="N",
                           help="DataLoader worker process count")),
    ("--skip-invalid-size-inputs-valid-test",
     dict(action="store_true",
          help="drop over/under-sized valid+test examples")),
    (("--batch-size", "--max-sentences"),
     dict(type=int, metavar="N", help="sentences per batch")),
    ("--required-batch-size-multiple",
     dict(default=1, type=int, metavar="N",
          help="round batch sizes to a multiple of this")),
    ("--data-buffer-size", dict(default=10, type=int, metavar="N",
                                help="batches to prefetch ahead")),
]

_DATASET_TRAIN_FLAGS = [
    ("--train-subset", dict(default="train", metavar="SPLIT",
                            help="split used for training")),
    ("--valid-subset", dict(default="valid", metavar="SPLIT",
                            help="comma-separated validation splits")),
    ("--validate-interval", dict(type=int, default=1, metavar="N",
                                 help="validate every N epochs")),
    ("--validate-interval-updates", dict(type=int, default=0, metavar="N",
                                         help="validate every N updates")),
    ("--validate-after-updates", dict(type=int, default=0, metavar="N",
                                      help="skip validation before this "
                                           "many updates")),
    ("--fixed-validation-seed", dict(default=None, type=int, metavar="N",
                                     help="fixed RNG seed for validation")),
    ("--disable-validation", dict(action="store_true",
                                  help="never validate")),
    ("--batch-size-valid", dict(type=int, metavar="N",
                                help="validation batch size "
                                     "(defaults to --batch-size)")),
    ("--max-valid-steps", dict(type=int, metavar="N",
                               help="cap on validation batches")),
    ("--curriculum", dict(default=0, type=int, metavar="N",
                          help="no batch shuffling for the first N epochs")),
]

_OPTIMIZATION_FLAGS = [
    (("--max-epoch", "--me"), dict(default=0, type=int, metavar="N",
                                   help="stop at this epoch")),
    (("--max-update", "--mu"), dict(default=0, type=int, metavar="N",
                                    help="stop at this update")),
    ("--stop-time-hours", dict(default=0, type=float, metavar="N",
                               help="stop after this much cumulative "
                                    "training time (if >0)")),
    ("--clip-norm", dict(default=0.0, type=float, metavar="NORM",
                         help="gradient clipping threshold")),
    ("--per-sample-clip-norm",
     dict(default=0.0, type=float, metavar="PNORM",
          help="per-sample gradient clipping threshold (requires "
               "--ddp-backend no_c10d and fp16/bf16)")),
    ("--update-freq", dict(default="1", metavar="N1,N2,...,N_K",
                           type=lambda uf: eval_str_list(uf, type=int),
                           help="gradient accumulation: N_i micro-batches "
                                "per update in epoch i")),
    (("--lr", "--learning-rate"),
     dict(default="0.25", type=lambda x: eval_str_list(x, type=float),
          metavar="LR_1,LR_2,...,LR_N",
          help="learning rate for the first N epochs (interpretation "
               "depends on --lr-scheduler)")),
    ("--stop-min-lr", dict(default=-1, type=float, metavar="LR",
                           help="stop once the LR reaches this floor")),
]

_CHECKPOINT_FLAGS = [
    ("--save-dir", dict(metavar="DIR", default="checkpoints",
                        help="checkpoint destination")),
    ("--tmp-save-dir", dict(metavar="DIR", default="./",
                            help="staging dir for checkpoint writes "
                                 "(async-copied into save-dir)")),
    ("--restore-file", dict(default="checkpoint_last.pt",
                            help="checkpoint to resume from "
                                 "(default: <save-dir>/checkpoint_last.pt")),
    ("--finetune-from-model",
     dict(default=None, type=str,
          help="load pretrained weights only; meters and lr scheduler "
               "start fresh")),
    ("--load-from-ema", dict(action="store_true",
                             help="take the EMA weights from the checkpoint "
                                  "as the model weights")),
    ("--reset-dataloader", dict(action="store_true",
                                help="ignore the checkpointed dataloader "
                                     "position")),
    ("--reset-lr-scheduler", dict(action="store_true",
                                  help="ignore the checkpointed lr-scheduler "
                                       "state")),
    ("--reset-meters", dict(action="store_true",
                            help="ignore the checkpointed meters")),
    ("--reset-optimizer", dict(action="store_true",
                               help="ignore the checkpointed optimizer "
                                    "state")),
    ("--optimizer-overrides", dict(default="{}", type=str, metavar="DICT",
                                   help="dict overriding optimizer args on "
                                        "checkpoint load")),
    ("--save-interval", dict(type=int, default=1, metavar="N",
                             help="checkpoint every N epochs")),
    ("--save-interval-updates", dict(type=int, default=0, metavar="N",
                                     help="checkpoint (and validate) every "
                                          "N updates")),
    ("--keep-interval-updates", dict(type=int, default=-1, metavar="N",
                                     help="retain the last N update "
                                          "checkpoints")),
    ("--keep-last-epochs", dict(type=int, default=-1, metavar="N",
                                help="retain the last N epoch checkpoints")),
    ("--keep-best-checkpoints", dict(type=int, default=-1, metavar="N",
                                     help="retain the best N checkpoints by "
                                          "metric")),
    ("--no-save", dict(action="store_true",
                       help="never write checkpoints")),
    ("--no-epoch-checkpoints", dict(action="store_true",
                                    help="only keep last and best")),
    ("--no-last-checkpoints", dict(action="store_true",
                                   help="skip checkpoint_last.pt")),
    ("--no-save-optimizer-state",
     dict(action="store_true",
          help="exclude optimizer state from checkpoints")),
    ("--best-checkpoint-metric", dict(type=str, default="loss",
                                      help='metric selecting "best" '
                                           "checkpoints")),
    ("--maximize-best-checkpoint-metric",
     dict(action="store_true",
          help='treat larger metric values as better for "best"')),
    ("--patience", dict(type=int, default=-1, metavar="N",
                        help="early-stop after N validations without "
                             "improvement (interacts with "
                             "--validate-interval)")),
    ("--checkpoint-suffix", dict(type=str, default="",
                                 help="suffix appended to checkpoint names")),
]


# ---------------------------------------------------------------------------
# parser construction
# ---------------------------------------------------------------------------


def get_training_parser(default_task="translation"):
    parser = get_parser("Trainer", default_task)
    add_dataset_args(parser, train=True)
    add_distributed_training_args(parser)
    add_model_args(parser)
    add_optimization_args(parser)
    add_checkpoint_args(parser)
    return parser


def get_parser(desc, default_task="test"):
    # the --user-dir plugin module must be imported BEFORE the registries
    # are consulted, so probe for it with a throwaway parser first
    probe = argparse.ArgumentParser(add_help=False, allow_abbrev=False)
    probe.add_argument("--user-dir", default=None)
    probe_args, _ = probe.parse_known_args()
    utils.import_user_module(probe_args)

    parser = argparse.ArgumentParser(allow_abbrev=False, description=desc)
    _install(parser, _COMMON_FLAGS)

    from unicore_amd.registry import REGISTRIES

    for registry_name, REGISTRY in REGISTRIES.items():
        parser.add_argument(
            "--" + registry_name.replace("_", "-"),
            default=REGISTRY["default"],
            choices=REGISTRY["registry"].keys(),
        )

    from unicore_amd.tasks import TASK_REGISTRY

    parser.add_argument("--task", metavar="TASK", default=default_task,
                        choices=TASK_REGISTRY.keys(), help="task")
    return parser


def add_dataset_args(parser, train=False, gen=False):
    group = parser.add_argument_group("Dataset and data loading")
    _install(group, _DATASET_FLAGS)
    if train:
        _install(group, _DATASET_TRAIN_FLAGS)
    return group


def add_distributed_training_args(parser, default_world_size=None):
    group = parser.add_argument_group("Distributed training")
    if default_world_size is None:
        default_world_size = max(1, torch.cuda.device_count())
    visible_gpus = max(1, torch.cuda.device_count())
    _install(group, [
        ("--distributed-world-size",
         dict(type=int, metavar="N", default=default_world_size,
              help="total GPU count across all nodes "
                   "(default: all visible GPUs)")),
        ("--distributed-rank", dict(default=0, type=int,
                                    help="rank of this worker")),
        ("--distributed-backend",
         dict(default="nccl", type=str,
              help='distributed backend ("nccl" is RCCL on ROCm)')),
        ("--distributed-init-method",
         dict(default=None, type=str,
              help="rendezvous address, typically tcp://hostname:port")),
        ("--distributed-port",
         dict(default=-1, type=int,
              help="rendezvous port (unneeded with "
                   "--distributed-init-method)")),
        (("--device-id", "--local_rank", "--local-rank"),
         dict(default=0, type=int,
              help="GPU index (usually configured automatically)")),
        ("--distributed-no-spawn",
         dict(action="store_true",
              help="never fork worker processes, even with multiple "
                   "visible GPUs")),
        ("--distributed-init-timeout",
         dict(default=300, type=int,
              help="init_process_group timeout in seconds")),
        ("--ddp-backend",
         dict(default="c10d", type=str,
              choices=["c10d", "pytorch_ddp", "flat", "no_c10d",
                       "legacy_ddp"],
              help="DistributedDataParallel engine: c10d/pytorch_ddp/flat "
                   "= FlatDDP (backward-overlapped bucketed all-reduce on "
                   "a side HIP stream); no_c10d/legacy_ddp = post-backward "
                   "bucketing")),
        ("--bucket-cap-mb",
         dict(default=32, type=int, metavar="MB",
              help="gradient all-reduce bucket size (tuned for xGMI)")),
        ("--fix-batches-to-gpus",
         dict(action="store_true",
              help="pin batches to fixed shards across epochs (less "
                   "randomness, avoids re-reading data)")),
        ("--find-unused-parameters",
         dict(default=False, action="store_true",
              help="tolerate params absent from the autograd graph "
                   "(not applicable to no_c10d)")),
        ("--broadcast-buffers",
         dict(default=False, action="store_true",
              help="sync non-trainable buffers (e.g. batchnorm stats) "
                   "across GPUs")),
        ("--nprocs-per-node",
         dict(type=int, metavar="N", default=visible_gpus,
              help="GPUs per node (intra-node all-reduce is cheap; "
                   "cross-node traffic is minimized)")),
    ])
    return group


def add_optimization_args(parser):
    group = parser.add_argument_group("Optimization")
    _install(group, _OPTIMIZATION_FLAGS)
    return group


def add_checkpoint_args(parser):
    group = parser.add_argument_group("Checkpointing")
    _install(group, _CHECKPOINT_FLAGS)
    return group


def add_model_args(parser):
    group = parser.add_argument_group("Model configuration")
    from unicore_amd.models import ARCH_MODEL_REGISTRY

    group.add_argument("--arch", "-a", default=None, metavar="ARCH",
                       choices=ARCH_MODEL_REGISTRY.keys() or None,
                       help="model architecture")
    return group


# ---------------------------------------------------------------------------
# two-pass parse
# ---------------------------------------------------------------------------


def parse_args_and_arch(
    parser: argparse.ArgumentParser,
    input_args: List[str] = None,
    parse_known: bool = False,
    suppress_defaults: bool = False,
    modify_parser: Optional[Callable[[argparse.ArgumentParser], None]] = None,
):
    """Parse twice: first to learn which plugins (model/task/loss/optimizer/
    lr_scheduler) are selected, then again after those classes have
    injected their own flags.

    *suppress_defaults* re-parses with all defaults nulled and returns only
    explicitly-set values; *modify_parser* runs before each pass.
    """
    if suppress_defaults:
        # learn the full flag surface first, then null every default
        seeded = parse_args_and_arch(
            parser, input_args=input_args, parse_known=parse_known,
            suppress_defaults=False,
        )
        nulled = argparse.ArgumentParser(add_help=False, parents=[parser])
        nulled.set_defaults(**{k: None for k in vars(seeded)})
        explicit = nulled.parse_args(input_args)
        return argparse.Namespace(
            **{k: v for k, v in vars(explicit).items() if v is not None}
        )

    from unicore_amd.models import (
        ARCH_CONFIG_REGISTRY,
        ARCH_MODEL_REGISTRY,
        MODEL_REGISTRY,
    )

    # the --user-dir plugins must load before their registrations are needed
    probe = argparse.ArgumentParser(add_help=False, allow_abbrev=False)
    probe.add_argument("--user-dir", default=None)
    probe_args, _ = probe.parse_known_args(input_args)
    utils.import_user_module(probe_args)

    if modify_parser is not None:
        modify_parser(parser)

    # pass 1: discover the selected plugin classes
    args, _ = parser.parse_known_args(input_args)

    if hasattr(args, "arch"):
        arch_group = parser.add_argument_group(
            "Model-specific configuration",
            # keep unset model flags out of the namespace
            argument_default=argparse.SUPPRESS,
        )
        if args.arch in ARCH_MODEL_REGISTRY:
            ARCH_MODEL_REGISTRY[args.arch].add_args(arch_group)
        elif args.arch in MODEL_REGISTRY:
            MODEL_REGISTRY[args.arch].add_args(arch_group)
        else:
            raise RuntimeError()

    from unicore_amd.registry import REGISTRIES

    for registry_name, REGISTRY in REGISTRIES.items():
        choice = getattr(args, registry_name, None)
        if choice is not None:
            plugin_cls = REGISTRY["registry"][choice]
            if hasattr(plugin_cls, "add_args"):
                plugin_cls.add_args(parser)
    if hasattr(args, "task"):
        from unicore_amd.tasks import TASK_REGISTRY

        TASK_REGISTRY[args.task].add_args(parser)

    # plugins may have reset defaults; let the caller adjust again
    if modify_parser is not None:
        modify_parser(parser)

    # pass 2: the real parse with the full flag surface
    extra = None
    if parse_known:
        args, extra = parser.parse_known_args(input_args)
    else:
        args = parser.parse_args(input_args)

    # post-processing / derived defaults
    if hasattr(args, "batch_size_valid") and args.batch_size_valid is None:
        args.batch_size_valid = args.batch_size
    args.bf16 = getattr(args, "bf16", False)
    args.bf16_sr = getattr(args, "bf16_sr", False)
    if getattr(args, "seed", None) is None:
        args.seed = 1  # default training seed
        args.no_seed_provided = True
    else:
        args.no_seed_provided = False

    if hasattr(args, "arch") and args.arch in ARCH_CONFIG_REGISTRY:
        ARCH_CONFIG_REGISTRY[args.arch](args)  # apply arch default table

    return (args, extra) if parse_known else args
