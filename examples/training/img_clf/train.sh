#!/bin/bash
# MNIST image classifier: 907K-param Perceiver IO with repeated cross-attention
# (latents 32x128, 2 CA layers, 3 SA blocks x 3 layers, 32 Fourier bands)
# (reference examples/training/img_clf/train.sh).
python -m perceiver_amd.scripts.vision.image_classifier fit \
  --model.num_latents 32 \
  --model.num_latent_channels 128 \
  --model.encoder.num_frequency_bands 32 \
  --model.encoder.num_cross_attention_layers 2 \
  --model.encoder.num_self_attention_layers_per_block 3 \
  --model.encoder.num_self_attention_blocks 3 \
  --model.encoder.first_cross_attention_layer_shared false \
  --model.encoder.first_self_attention_block_shared true \
  --model.encoder.dropout 0.0 \
  --model.decoder.num_output_query_channels 128 \
  --model.decoder.dropout 0.0 \
  --data.batch_size 128 \
  --data.random_crop 28 \
  --optimizer.lr 1e-3 \
  --trainer.max_epochs 20 \
  --trainer.out_dir logs/img_clf
