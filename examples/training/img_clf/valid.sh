#!/bin/bash
# Validate from a checkpoint (reference examples/training/img_clf/valid.sh).
python -m perceiver_amd.scripts.vision.image_classifier validate \
  --config logs/img_clf/config.yaml \
  --trainer.ckpt_path logs/img_clf/checkpoints/best.ckpt
