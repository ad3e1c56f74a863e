"""Mask-filling helper for qualitative MLM evaluation.

Turns ``<mask>`` placeholders in plain text into the tokenizer's mask token,
runs the model once, and decodes the top-k replacement candidates per masked
position — the table logged at validation end mirrors the reference's
text/mlm/utils.py behavior.
"""
from __future__ import annotations

import torch


class MaskFiller:
    """Fill ``<mask>`` slots in a batch of strings with model predictions."""

    def __init__(self, preprocessor):
        self.preprocessor = preprocessor

    @property
    def _tokenizer(self):
        return self.preprocessor.tokenizer

    def fill(self, model, masked_text_batch, num_predictions, device="cpu"):
        """Returns (normalized inputs, per-sample list of k filled strings)."""
        texts = [s.replace("<mask>", self._tokenizer.mask_token) for s in masked_text_batch]
        token_ids, pad_mask = self.preprocessor.preprocess_batch(texts)
        token_ids = token_ids.to(device)

        with torch.no_grad():
            logits = model(token_ids, pad_mask.to(device))

        at_mask = token_ids == self._tokenizer.mask_token_id
        candidates = torch.topk(logits[at_mask, :], k=num_predictions, dim=1).indices

        # decode once per rank, substituting every masked position in place
        decoded_per_rank = []
        for rank in range(num_predictions):
            token_ids[at_mask] = candidates[:, rank]
            decoded_per_rank.append(self._tokenizer.batch_decode(token_ids, skip_special_tokens=True))

        per_sample = [list(ranks) for ranks in zip(*decoded_per_rank)]
        return texts, per_sample
