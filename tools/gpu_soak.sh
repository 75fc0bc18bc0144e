set -x
cd /root/repo
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -m unicore_cli.train \
  --task bert_synthetic --arch bert_base --loss masked_lm \
  --optimizer adam --adam-betas '(0.9, 0.98)' --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 3e-4 --warmup-updates 50 \
  --total-num-update 100000 --max-update 200 --dataset-size 16384 \
  --batch-size 64 --tokens-per-sample 512 --max-seq-len 514 --vocab-size 30522 \
  --bf16 --ddp-backend c10d --log-interval 40 --log-format simple --no-save \
  --save-dir /tmp/ck_soak 2>&1 | grep -E "train_inner" | tail -5
timeout 240 python -m unicore_cli.train \
  --task unimol_synthetic --arch mol_pairbias --loss mol_pretrain \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 1.0 \
  --lr-scheduler polynomial_decay --lr 3e-4 --warmup-updates 50 \
  --total-num-update 50000 --max-update 150 --dataset-size 4800 \
  --batch-size 32 --atoms-per-mol 256 --bf16 --ddp-backend c10d \
  --log-interval 50 --log-format simple --no-save \
  --save-dir /tmp/ck_soak2 2>&1 | grep -E "train_inner" | tail -3
timeout 240 python -m unicore_cli.train \
  --task evoformer_synthetic --arch evoformer --loss masked_msa \
  --optimizer adam --adam-betas '(0.9, 0.99)' --adam-eps 1e-6 --clip-norm 0.1 \
  --lr-scheduler polynomial_decay --lr 1e-3 --warmup-updates 50 \
  --total-num-update 20000 --max-update 48 --dataset-size 512 \
  --batch-size 1 --update-freq 8 --msa-depth 128 --residues 256 \
  --bf16 --bf16-sr --ddp-backend c10d --log-interval 16 --log-format simple \
  --no-save --save-dir /tmp/ck_soak3 2>&1 | grep -E "train_inner" | tail -3
