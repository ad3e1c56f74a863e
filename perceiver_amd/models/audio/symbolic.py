"""Symbolic audio model = CausalSequenceModel (Perceiver AR) over MIDI event tokens.

Parity: /root/reference/perceiver/model/audio/symbolic/backend.py:11-13.
"""
from __future__ import annotations

from dataclasses import dataclass

from perceiver_amd.core import CausalSequenceModel, CausalSequenceModelConfig


@dataclass
class SymbolicAudioModelConfig(CausalSequenceModelConfig):
    pass


class SymbolicAudioModel(CausalSequenceModel):
    def __init__(self, config: SymbolicAudioModelConfig):
        super().__init__(config)
