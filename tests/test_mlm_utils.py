"""MaskFiller behavior (reference text/mlm/utils.py + mask_filler_test.py
category 7: a mock model with rigged logits makes predictions deterministic)."""
import torch

from perceiver_amd.data.text.common import TextPreprocessor
from perceiver_amd.models.text.mlm_utils import MaskFiller


class RiggedModel:
    """Returns logits that put probability mass on token id = (position index
    of the mask + base), making top-k predictions fully predictable."""

    def __init__(self, vocab_size, base):
        self.vocab_size = vocab_size
        self.base = base

    def __call__(self, xs, pad_mask):
        b, n = xs.shape
        logits = torch.zeros(b, n, self.vocab_size)
        # rank tokens base, base+1, base+2, ... descending at every position
        for k in range(5):
            logits[:, :, self.base + k] = 5.0 - k
        return logits


def _preproc():
    from transformers import PerceiverTokenizer

    tok = PerceiverTokenizer()
    return TextPreprocessor(tokenizer=tok, max_seq_len=32, add_special_tokens=False)


def test_mask_filler_topk_predictions():
    pre = _preproc()
    tok = pre.tokenizer
    model = RiggedModel(vocab_size=tok.vocab_size, base=tok("k", add_special_tokens=False)["input_ids"][0])
    filler = MaskFiller(pre)

    masked, results = filler.fill(model, ["a <mask> b", "c d <mask>"], num_predictions=2)
    assert len(masked) == 2 and len(results) == 2
    assert all(len(r) == 2 for r in results)
    # the mask token was substituted: outputs are plain strings without <mask>
    for r in results:
        for s in r:
            assert "<mask>" not in s and isinstance(s, str)
    # rigged logits: top-1 fills 'k', top-2 fills the next byte token
    assert results[0][0] == "a k b"
    assert results[1][0] == "c d k"
