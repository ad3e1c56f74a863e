#!/bin/bash
# IMDb sentiment, full fine-tune (encoder unfrozen)
# (reference examples/training/txt_clf/train_all.sh).
python -m perceiver_amd.scripts.text.classifier fit \
  --model.params logs/txt_clf_dec/checkpoints/best.ckpt \
  --data.batch_size 32 \
  --optimizer.lr 2e-5 \
  --trainer.max_epochs 4 \
  --trainer.out_dir logs/txt_clf_all
