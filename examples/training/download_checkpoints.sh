#!/bin/bash
# Mirror of the reference's published Lightning checkpoints (for conversion and
# validation runs). Requires network access.
set -e
BASE="https://martin-krasser.com/perceiver/logs-0.8.0"
DEST="${1:-logs}"
mkdir -p "$DEST"
for run in mlm txt_clf img_clf clm sam; do
  echo "fetching $run checkpoints into $DEST/$run ..."
  mkdir -p "$DEST/$run/checkpoints"
  wget -q -r -np -nH --cut-dirs=2 -P "$DEST/$run/checkpoints" "$BASE/$run/checkpoints/" || true
done
