set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 900 python -m pytest tests/ -q -m gpu > gpurun_out/full_gpu.log 2>&1
echo SUITE_RC=$?; grep -E 'passed|failed' gpurun_out/full_gpu.log | tail -1
timeout 200 python -c 'import __graft_entry__ as g; g.smoke(); print("SMOKE OK")' 2>&1 | tail -1
timeout 300 python bench.py --steps 20 --warmup 5 2>/dev/null | grep -o 'value": [0-9.]*'
# recycling + activation checkpointing on GPU bf16
timeout 300 python -m unicore_cli.train \
  --task evoformer_synthetic --arch evoformer --loss masked_msa \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
  --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 100 \
  --total-num-update 20000 --max-update 8 --dataset-size 64 \
  --batch-size 1 --update-freq 4 --msa-depth 64 --residues 128 \
  --recycle-iters 1 --activation-checkpoint \
  --bf16 --bf16-sr --ddp-backend c10d --log-interval 4 --log-format simple \
  --no-save --save-dir /tmp/ck_r 2>&1 | grep train_inner | tail -1
# train-save-infer on GPU with flash under no_grad
timeout 300 python -m unicore_cli.train \
  --task bert_synthetic --arch bert_base --loss masked_lm \
  --optimizer adam --lr-scheduler fixed --lr 1e-4 --max-update 3 \
  --dataset-size 16 --batch-size 4 --tokens-per-sample 512 --max-seq-len 514 \
  --vocab-size 30522 --bf16 --ddp-backend c10d --log-format simple \
  --num-workers 0 --save-dir /tmp/ck_inf > /dev/null 2>&1
timeout 200 python examples/bert/infer_demo.py --checkpoint /tmp/ck_inf/checkpoint_last.pt --bf16 --batch-size 32 --seq-len 512 --iters 5 2>&1 | tail -1
