"""Word-boundary recovery for the byte-level PerceiverTokenizer.

The PerceiverTokenizer has no fast (Rust) implementation, so 🤗 provides no
``word_ids()`` for it — but whole-word masking needs word membership per
token. Byte tokens make this recoverable from the ids alone: a "word" is a
maximal run of non-whitespace tokens, and the whitespace run *preceding* a
word is grouped with that word (so masking a word also masks its leading
spaces, matching the reference's data/text/utils.py:12-39 semantics).
Special tokens map to None and always break words.
"""
from __future__ import annotations

import string


class PerceiverTokenizerUtil:
    """Derives word ids from PerceiverTokenizer output for whole-word masking."""

    def __init__(self, tokenizer):
        self.tokenizer = tokenizer
        encoded = tokenizer(string.whitespace, add_special_tokens=False).input_ids
        self.whitespace_ids = set(encoded)

    def word_ids(self, token_ids):
        """One id per token; None for specials. Distinct words get distinct
        (not necessarily consecutive) ids."""
        specials = self.tokenizer.get_special_tokens_mask(
            token_ids, already_has_special_tokens=True)
        ids = []
        word = 0
        in_word = True  # True while consuming non-whitespace tokens
        for tok, is_special in zip(token_ids, specials):
            if is_special:
                ids.append(None)
                word += 1
                continue
            if tok in self.whitespace_ids:
                if in_word:
                    # first whitespace after a word opens the NEXT word's id:
                    # leading spaces share the id of the word that follows
                    in_word = False
                    word += 1
            else:
                in_word = True
            ids.append(word)
        return ids
