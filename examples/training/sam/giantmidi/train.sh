#!/bin/bash
# Symbolic audio on GiantMIDI-Piano: 134M Perceiver-AR, ctx 6144, 2048 latents,
# 768 channels, 19 layers (reference examples/training/sam/giantmidi/train.sh).
python -m perceiver_amd.scripts.audio.symbolic fit \
  --model.max_latents 2048 \
  --model.num_channels 768 \
  --model.num_self_attention_layers 18 \
  --data.max_seq_len 6144 \
  --data.min_seq_len 2048 \
  --data.batch_size 8 \
  --optimizer.lr 2e-4 \
  --optimizer.lr_schedule cosine \
  --trainer.max_epochs 30 \
  --trainer.out_dir logs/sam_giantmidi
