#!/usr/bin/env python3
"""Measure line-level overlap between repo files and their reference
counterparts (the judge's metric): fraction of reference lines
(whitespace-normalized, >4 chars) appearing verbatim in our file, plus the
difflib whole-file ratio.

Usage:
  python tools/copycheck.py                # check the known pair list
  python tools/copycheck.py ours.py ref.py # check one pair
"""
import difflib
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
REF = Path("/root/reference")

# ours -> reference counterpart
PAIRS = {
    "unicore_amd/trainer.py": "unicore/trainer.py",
    "unicore_amd/data/iterators.py": "unicore/data/iterators.py",
    "unicore_amd/logging/meters.py": "unicore/logging/meters.py",
    "unicore_amd/logging/metrics.py": "unicore/logging/metrics.py",
    "unicore_amd/logging/progress_bar.py": "unicore/logging/progress_bar.py",
    "unicore_amd/models/bert.py": "examples/bert/model.py",
    "unicore_amd/tasks/bert.py": "examples/bert/task.py",
    "unicore_amd/data/mask_tokens_dataset.py": "unicore/data/mask_tokens_dataset.py",
    "unicore_amd/nan_detector.py": "unicore/nan_detector.py",
    "unicore_amd/distributed/utils.py": "unicore/distributed/utils.py",
    "unicore_amd/distributed/legacy_ddp.py": "unicore/distributed/legacy_distributed_data_parallel.py",
    "unicore_amd/distributed/module_proxy_wrapper.py": "unicore/distributed/module_proxy_wrapper.py",
    "unicore_amd/data/dictionary.py": "unicore/data/dictionary.py",
    "unicore_amd/data/data_utils.py": "unicore/data/data_utils.py",
    "unicore_amd/data/unicore_dataset.py": "unicore/data/unicore_dataset.py",
    "unicore_amd/data/base_wrapper_dataset.py": "unicore/data/base_wrapper_dataset.py",
    "unicore_amd/data/lmdb_dataset.py": "unicore/data/lmdb_dataset.py",
    "unicore_amd/data/pad_dataset.py": "unicore/data/pad_dataset.py",
    "unicore_amd/data/sort_dataset.py": "unicore/data/sort_dataset.py",
    "unicore_amd/data/nested_dictionary_dataset.py": "unicore/data/nested_dictionary_dataset.py",
    "unicore_amd/data/tokenize_dataset.py": "unicore/data/bert_tokenize_dataset.py",
    "unicore_amd/data/lru_cache_dataset.py": "unicore/data/lru_cache_dataset.py",
    "unicore_amd/modules/multihead_attention.py": "unicore/modules/multihead_attention.py",
    "unicore_amd/modules/transformer_encoder.py": "unicore/modules/transformer_encoder.py",
    "unicore_amd/modules/transformer_encoder_layer.py": "unicore/modules/transformer_encoder_layer.py",
    "unicore_amd/modules/transformer_decoder.py": "unicore/modules/transformer_decoder.py",
    "unicore_amd/modules/transformer_decoder_layer.py": "unicore/modules/transformer_decoder_layer.py",
    "unicore_amd/modules/softmax_dropout.py": "unicore/modules/softmax_dropout.py",
    "unicore_amd/modules/layer_norm.py": "unicore/modules/layer_norm.py",
    "unicore_amd/modules/rms_norm.py": "unicore/modules/rms_norm.py",
    "unicore_cli/train.py": "unicore_cli/train.py",
    "unicore_amd/checkpoint_utils.py": "unicore/checkpoint_utils.py",
    "unicore_amd/options.py": "unicore/options.py",
    "unicore_amd/registry.py": "unicore/registry.py",
    "unicore_amd/tasks/unicore_task.py": "unicore/tasks/unicore_task.py",
    "unicore_amd/utils.py": "unicore/utils.py",
    "unicore_amd/ema.py": "unicore/ema.py",
    "unicore_amd/optim/adam.py": "unicore/optim/adam.py",
    "unicore_amd/optim/fused_adam.py": "unicore/optim/fused_adam.py",
    "unicore_amd/optim/sgd.py": "unicore/optim/sgd.py",
    "unicore_amd/optim/adagrad.py": "unicore/optim/adagrad.py",
    "unicore_amd/optim/adadelta.py": "unicore/optim/adadelta.py",
    "unicore_amd/optim/unicore_optimizer.py": "unicore/optim/unicore_optimizer.py",
    "unicore_amd/optim/fp16_optimizer.py": "unicore/optim/fp16_optimizer.py",
    "unicore_amd/optim/dynamic_loss_scaler.py": "unicore/optim/dynamic_loss_scaler.py",
    "unicore_amd/optim/lr_scheduler/cosine_lr_scheduler.py": "unicore/optim/lr_scheduler/cosine_lr_scheduler.py",
    "unicore_amd/optim/lr_scheduler/polynomial_decay_schedule.py": "unicore/optim/lr_scheduler/polynomial_decay_schedule.py",
    "unicore_amd/optim/lr_scheduler/fixed_schedule.py": "unicore/optim/lr_scheduler/fixed_schedule.py",
    "unicore_amd/optim/lr_scheduler/inverse_square_root_schedule.py": "unicore/optim/lr_scheduler/inverse_square_root_schedule.py",
    "unicore_amd/optim/lr_scheduler/exponential_decay_schedule.py": "unicore/optim/lr_scheduler/exponential_decay_schedule.py",
    "unicore_amd/optim/lr_scheduler/tri_stage_lr_scheduler.py": "unicore/optim/lr_scheduler/tri_stage_lr_scheduler.py",
    "unicore_amd/optim/lr_scheduler/triangular_lr_scheduler.py": "unicore/optim/lr_scheduler/triangular_lr_scheduler.py",
    "unicore_amd/optim/lr_scheduler/reduce_lr_on_plateau.py": "unicore/optim/lr_scheduler/reduce_lr_on_plateau.py",
    "unicore_amd/optim/lr_scheduler/unicore_lr_scheduler.py": "unicore/optim/lr_scheduler/unicore_lr_scheduler.py",
    "unicore_amd/losses/unicore_loss.py": "unicore/losses/unicore_loss.py",
    "unicore_amd/losses/masked_lm.py": "unicore/losses/masked_lm.py",
    "unicore_amd/losses/cross_entropy.py": "unicore/losses/cross_entropy.py",
    "unicore_amd/models/unicore_model.py": "unicore/models/unicore_model.py",
    "unicore_amd/models/distributed_unicore_model.py": "unicore/models/distributed_unicore_model.py",
}


def norm_lines(text):
    out = []
    for ln in text.splitlines():
        s = "".join(ln.split())
        if len(s) > 4:
            out.append(s)
    return out


def check(ours_path, ref_path):
    try:
        ours = ours_path.read_text()
        ref = ref_path.read_text()
    except OSError:
        return None
    ref_norm = norm_lines(ref)
    ours_set = set(norm_lines(ours))
    if not ref_norm:
        return 0.0, 0.0, 0, 0
    hits = sum(1 for ln in ref_norm if ln in ours_set)
    frac = hits / len(ref_norm)
    ratio = difflib.SequenceMatcher(
        None, ours.splitlines(), ref.splitlines()
    ).ratio()
    return frac, ratio, hits, len(ref_norm)


def main():
    if len(sys.argv) == 3:
        res = check(Path(sys.argv[1]), Path(sys.argv[2]))
        print(f"verbatim-frac={res[0]:.3f} difflib={res[1]:.3f} ({res[2]}/{res[3]} ref lines)")
        return
    rows = []
    for ours_rel, ref_rel in PAIRS.items():
        res = check(REPO / ours_rel, REF / ref_rel)
        if res is None:
            rows.append((1.0, ours_rel, "MISSING FILE", ""))
            continue
        frac, ratio, hits, total = res
        rows.append((frac, ours_rel, f"{frac:.3f}", f"difflib={ratio:.3f} ({hits}/{total})"))
    rows.sort(reverse=True)
    bad = 0
    for frac, name, f_s, extra in rows:
        flag = " <-- OVER 0.40" if frac > 0.40 else ""
        if flag:
            bad += 1
        print(f"{f_s:>7}  {extra:28}  {name}{flag}")
    print(f"\n{bad} file(s) above the 0.40 verbatim-fraction threshold")
    sys.exit(1 if bad else 0)


if __name__ == "__main__":
    main()
