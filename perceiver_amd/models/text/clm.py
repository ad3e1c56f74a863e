"""Causal language model = CausalSequenceModel (Perceiver AR).

Parity: /root/reference/perceiver/model/text/clm/backend.py:11-13.
"""
from __future__ import annotations

from dataclasses import dataclass

from perceiver_amd.core import CausalSequenceModel, CausalSequenceModelConfig


@dataclass
class CausalLanguageModelConfig(CausalSequenceModelConfig):
    pass


class CausalLanguageModel(CausalSequenceModel):
    def __init__(self, config: CausalLanguageModelConfig):
        super().__init__(config)
