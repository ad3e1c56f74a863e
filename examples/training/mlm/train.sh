#!/bin/bash
# Masked LM fine-tuning on IMDb (201M Perceiver IO, UTF-8 bytes, seq 2048)
# — the flagship MLM configuration (reference examples/training/mlm/train.sh).
python -m perceiver_amd.scripts.text.mlm fit \
  --data.batch_size 32 \
  --data.max_seq_len 2048 \
  --optimizer.lr 2e-4 \
  --optimizer.warmup_steps 1000 \
  --trainer.max_epochs 12 \
  --trainer.precision bf16 \
  --trainer.out_dir logs/mlm
