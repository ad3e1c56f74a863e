#!/bin/bash
# IMDb sentiment, decoder-only training on a frozen 201M MLM encoder
# (reference examples/training/txt_clf/train_dec.sh).
python -m perceiver_amd.scripts.text.classifier fit \
  --model.encoder.freeze true \
  --model.encoder.params logs/mlm/checkpoints/best.ckpt \
  --data.batch_size 64 \
  --optimizer.lr 1e-3 \
  --trainer.max_epochs 6 \
  --trainer.out_dir logs/txt_clf_dec
