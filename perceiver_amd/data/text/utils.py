"""Support for the (non-fast) PerceiverTokenizer: whitespace-boundary word ids for
whole-word masking (parity with reference data/text/utils.py:12-39)."""
from __future__ import annotations

import string


class PerceiverTokenizerUtil:
    def __init__(self, tokenizer):
        self.tokenizer = tokenizer
        self.whitespace_ids = set(tokenizer(string.whitespace, add_special_tokens=False).input_ids)

    def word_ids(self, token_ids):
        """Word ids from token ids using whitespace boundaries. Whitespaces preceding
        a word share its word id; special tokens get None; distinct words are
        guaranteed distinct ids (not necessarily consecutive)."""
        word_ids = []
        curr_id = 0
        special_mask = self.tokenizer.get_special_tokens_mask(token_ids, already_has_special_tokens=True)
        regular_token = True
        for i, token_id in enumerate(token_ids):
            if special_mask[i]:
                word_ids.append(None)
                curr_id += 1
            elif token_id in self.whitespace_ids:
                if regular_token:
                    regular_token = False
                    curr_id += 1
                word_ids.append(curr_id)
            else:
                regular_token = True
                word_ids.append(curr_id)
        return word_ids
