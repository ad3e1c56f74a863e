#!/bin/bash
# 455M Perceiver-AR on C4 (streaming), FSDP-sharded: ctx 1024, 512 latents,
# 1280 channels, 20+1 layers, 32k SentencePiece vocab
# (reference examples/training/clm/train_fsdp.sh; on MI355X's 288 GB HBM plain DDP
# also fits this model — FSDP kept for much larger configs).
python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \
  -m perceiver_amd.scripts.text.clm_fsdp fit \
  --data.batch_size 32 \
  --data.max_seq_len 1024 \
  --data.min_seq_len 512 \
  --optimizer.lr 2e-4 \
  --optimizer.lr_schedule cosine \
  --optimizer.warmup_steps 1000 \
  --trainer.max_steps 50000 \
  --trainer.grad_clip 0.5 \
  --trainer.out_dir logs/clm_fsdp
