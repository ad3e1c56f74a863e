"""Symbolic-audio (MIDI event token) data pipeline.

Preprocessing encodes every MIDI file into the 388-event token alphabet
(data/audio/midi_processor.py) and concatenates all pieces into ONE flat
int16 stream per split, with ``-1`` separators between pieces, persisted as a
np.memmap ``.bin``. Training samples are drawn by random position: a window
of ``max_seq_len + 1`` tokens is cut anywhere in the stream, the longest
separator-free span inside it is kept, and the collator pads (pad id 388,
vocab 389) and shifts by one into (labels, input_ids, pad_mask).

Behavior mirrors the reference's data/audio/symbolic.py:16-232; subclasses
provide ``load_source_dataset`` (see maestro_v3.py / giantmidi_piano.py).
"""
from __future__ import annotations

import functools
import os
import random
from pathlib import Path
from typing import Dict, List, Optional

import numpy as np
import torch
import torch.nn.functional as F
from torch.utils.data import DataLoader

from perceiver_amd.data.audio.midi_processor import encode_midi_files
from perceiver_amd.data.text.common import Hparams

SEPARATOR_ID = -1   # piece boundary in the flat stream (never a real token)
PAD_ID = 388        # one past the 0..387 event alphabet
VOCAB_SIZE = 389


def _flatten_with_separators(pieces: List[np.ndarray]) -> np.ndarray:
    """All pieces -> one stream, a separator after each piece."""
    sep = np.array([SEPARATOR_ID], dtype=pieces[0].dtype if pieces else np.int16)
    return np.concatenate([np.concatenate([ids, sep]) for ids in pieces])


def _write_memmap(stream: np.ndarray, path: Path) -> None:
    sink = np.memmap(str(path.absolute()), dtype=np.int16, mode="w+", shape=stream.shape)
    sink[:] = stream[:]
    sink.flush()


class SymbolicAudioDataModule:
    """Base data module over a preprocessed MIDI token stream."""

    # kept as class attributes for API parity with subclass expectations
    _EXAMPLE_SEPARATOR_INPUT_ID = SEPARATOR_ID
    _PAD_INPUT_ID = PAD_ID
    _VOCAB_SIZE = VOCAB_SIZE

    def __init__(
        self,
        dataset_dir: str,
        max_seq_len: int,
        min_seq_len: Optional[int] = None,
        padding_side: str = "left",
        batch_size: int = 16,
        num_workers: int = 1,
        preproc_workers: Optional[int] = None,
        pin_memory: bool = True,
    ):
        if min_seq_len is not None and not (0 < min_seq_len < max_seq_len):
            raise ValueError(
                "Invalid data configuration supplied. "
                "Parameter 'min_seq_len' must adhere to 0 < min_seq_len < max_seq_len."
            )
        self.hparams = Hparams(dataset_dir=dataset_dir, max_seq_len=max_seq_len,
                               min_seq_len=min_seq_len, padding_side=padding_side,
                               batch_size=batch_size, num_workers=num_workers,
                               preproc_workers=preproc_workers, pin_memory=pin_memory)
        # windows carry one extra token: the collator consumes it for the shift
        self._collator = SymbolicAudioCollator(
            max_seq_len=max_seq_len + 1, pad_token=PAD_ID, padding_side=padding_side,
        )
        self._ds_train: Optional[SymbolicAudioNumpyDataset] = None
        self._ds_valid: Optional[SymbolicAudioNumpyDataset] = None

    # ---------------------------------------------------------------- schema
    @property
    def vocab_size(self):
        return VOCAB_SIZE

    @property
    def max_seq_len(self):
        return self.hparams.max_seq_len

    @property
    def preproc_workers(self):
        declared = self.hparams.preproc_workers
        return declared if declared is not None else max(1, self.hparams.num_workers)

    @property
    @functools.lru_cache(maxsize=1)
    def preproc_dir(self) -> Path:
        return Path(self.hparams.dataset_dir) / "preproc"

    @property
    def train_data_file(self) -> Path:
        return self.preproc_dir / "train.bin"

    @property
    def valid_data_file(self) -> Path:
        return self.preproc_dir / "valid.bin"

    # ----------------------------------------------------------- preparation
    def load_source_dataset(self) -> Dict[str, Path]:
        """Subclass hook: {'train': dir, 'valid': dir} of MIDI directories."""
        raise NotImplementedError

    def _encode_split(self, midi_dir: Path) -> List[np.ndarray]:
        midi_dir = Path(midi_dir)
        if not midi_dir.exists():
            raise ValueError(f"Invalid directory supplied. Directory '{midi_dir}' does not exist.")
        found = [*midi_dir.rglob("**/*.mid"), *midi_dir.rglob("**/*.midi")]
        return encode_midi_files(found, num_workers=self.preproc_workers)

    def prepare_data(self) -> None:
        if os.path.exists(self.preproc_dir):
            return
        sources = self.load_source_dataset()
        train_pieces = self._encode_split(sources["train"])
        valid_pieces = self._encode_split(sources["valid"])
        random.shuffle(train_pieces)
        self.preproc_dir.mkdir(parents=True)
        _write_memmap(_flatten_with_separators(train_pieces), self.train_data_file)
        _write_memmap(_flatten_with_separators(valid_pieces), self.valid_data_file)

    # -------------------------------------------------------------- loading
    def setup(self, stage: Optional[str] = None) -> None:
        window = self.hparams.max_seq_len + 1
        min_len = self.hparams.min_seq_len
        self._ds_train = SymbolicAudioNumpyDataset(
            data_file=str(self.train_data_file), max_seq_len=window,
            separator_input_id=SEPARATOR_ID,
            min_seq_len=min_len + 1 if min_len is not None else None,
        )
        self._ds_valid = SymbolicAudioNumpyDataset(
            data_file=str(self.valid_data_file), max_seq_len=window,
            separator_input_id=SEPARATOR_ID,
        )

    def _loader(self, dataset) -> DataLoader:
        return DataLoader(dataset, shuffle=False, collate_fn=self._collator,
                          batch_size=self.hparams.batch_size,
                          num_workers=self.hparams.num_workers,
                          pin_memory=self.hparams.pin_memory)

    def train_dataloader(self):
        return self._loader(self._ds_train)

    def val_dataloader(self):
        return self._loader(self._ds_valid)


class SymbolicAudioNumpyDataset(torch.utils.data.Dataset):
    """Random-position window sampling over the flat memmap stream.

    A window may straddle piece boundaries; the longest separator-free span
    wins (so examples never mix pieces). With ``min_seq_len`` set, examples
    are additionally truncated to a random length in [min, max) — variable-
    length training for robust short-prompt generation.
    """

    def __init__(self, data_file: str, max_seq_len: int, separator_input_id: int,
                 min_seq_len: Optional[int] = None):
        self._data = np.memmap(data_file, dtype=np.int16, mode="r")
        self._max_seq_len = max_seq_len
        self._separator_input_id = separator_input_id
        self._min_seq_len = min_seq_len

    def __len__(self):
        # nominal epoch size: non-overlapping window count
        return self._data.shape[0] // self._max_seq_len

    def _longest_clean_span(self, window: torch.Tensor) -> torch.Tensor:
        boundaries = torch.where(window == self._separator_input_id)[0]
        if boundaries.numel() == 0:
            return window
        spans = torch.tensor_split(window, boundaries)
        best = max(spans, key=len)
        return best[best != self._separator_input_id]

    def __getitem__(self, index):
        top = self._data.shape[0] - self._max_seq_len
        at = torch.randint(top, (1,)).item()
        window = torch.tensor(self._data[at: at + self._max_seq_len].astype(np.int64))
        example = self._longest_clean_span(window)
        if self._min_seq_len is not None and self._min_seq_len < len(example):
            keep = torch.randint(self._min_seq_len, self._max_seq_len, (1,)).item()
            example = example[:keep]
        return {"input_ids": example}


class SymbolicAudioCollator:
    """Pad to the window length (left or right), then shift by one:
    labels = window[1:], inputs = window[:-1], pad_mask True at padding."""

    def __init__(self, max_seq_len: int, pad_token: int, padding_side: str):
        if padding_side not in ("left", "right"):
            raise ValueError(f"Invalid padding side '{padding_side}'")
        self._max_seq_len = max_seq_len
        self._pad_token = pad_token
        self._padding_side = padding_side

    def _pad_one(self, ids: torch.Tensor):
        missing = self._max_seq_len - len(ids)
        if missing == 0:
            return ids, torch.zeros(ids.shape)
        edges = (missing, 0) if self._padding_side == "left" else (0, missing)
        padded = F.pad(ids, edges, "constant", self._pad_token)
        mask = torch.zeros(padded.shape).masked_fill(padded == self._pad_token, 1)
        return padded, mask

    def __call__(self, input_batch):
        padded, masks = zip(*(self._pad_one(ex["input_ids"]) for ex in input_batch))
        windows = torch.stack(list(padded), dim=0)
        pad_mask = torch.stack(list(masks), dim=0)[..., :-1].type(torch.bool)
        return windows[..., 1:], windows[..., :-1], pad_mask
