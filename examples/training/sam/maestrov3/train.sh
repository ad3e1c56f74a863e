#!/bin/bash
# Symbolic audio on Maestro V3: 28.5M Perceiver-AR, ctx 2048, 1024 latents,
# 512 channels, 9 layers (reference examples/training/sam/maestrov3/train.sh).
python -m perceiver_amd.scripts.audio.symbolic fit \
  --model.max_latents 1024 \
  --model.num_channels 512 \
  --model.num_self_attention_layers 8 \
  --data.max_seq_len 2048 \
  --data.batch_size 48 \
  --optimizer.lr 2e-4 \
  --optimizer.lr_schedule cosine \
  --trainer.max_epochs 30 \
  --trainer.out_dir logs/sam_maestro
