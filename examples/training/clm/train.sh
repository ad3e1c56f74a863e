#!/bin/bash
# Causal LM on WikiText-103-raw bytes: 30.7M Perceiver-AR, ctx 4096, 512 latents,
# 512 channels, 8+1 layers, cross-attention dropout 0.5
# (reference examples/training/clm/train.sh).
python -m perceiver_amd.scripts.text.clm fit \
  --model.max_latents 512 \
  --model.num_channels 512 \
  --model.num_self_attention_layers 8 \
  --model.cross_attention_dropout 0.5 \
  --data.batch_size 24 \
  --data.max_seq_len 4096 \
  --data.random_train_shift true \
  --trainer.accumulate_grad_batches 2 \
  --optimizer.lr 2e-4 \
  --trainer.grad_clip 0.5 \
  --trainer.max_epochs 12 \
  --trainer.out_dir logs/clm
