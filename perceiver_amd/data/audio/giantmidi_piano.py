"""GiantMIDI-Piano symbolic-audio data module.

Unlike Maestro (see maestro_v3.py), the GiantMIDI archive already contains
`train/` and `valid/` directories, so loading reduces to download + extract +
pointing `SymbolicAudioDataModule` at the extracted split directories.
Reference behavior: data/audio/giantmidi_piano.py.
"""
from __future__ import annotations

import os
import shutil
from pathlib import Path
from typing import Any, Dict

from perceiver_amd.data.audio.symbolic import SymbolicAudioDataModule
from perceiver_amd.data.audio.utils import download_file, extract_file

_ZIP_NAME = "giantmidi-piano.zip"
_DEFAULT_URI = f"https://martin-krasser.com/perceiver/data/midi/{_ZIP_NAME}"
_SPLIT_ERRORS = {
    "train": "Could not find training directory in downloaded dataset (expected=`{}`)",
    "valid": "Could not find validation directory in downloaded dataset (expected=`{}`)",
}


class GiantMidiPianoDataModule(SymbolicAudioDataModule):
    """Symbolic-audio training on the GiantMIDI-Piano transcription corpus."""

    def __init__(
        self,
        *args: Any,
        dataset_uri: str = _DEFAULT_URI,
        dataset_dir: str = os.path.join(".cache", "giantmidi-piano"),
        **kwargs: Any,
    ):
        super().__init__(dataset_dir, *args, **kwargs)
        self._dataset_uri = dataset_uri

    @property
    def source_dir(self) -> Path:
        return Path(self.hparams.dataset_dir) / "source"

    def load_source_dataset(self) -> Dict[str, Path]:
        root = self.source_dir
        if root.exists():
            shutil.rmtree(root)
        scratch = root / "_download"
        scratch.mkdir(parents=True)

        archive = scratch / _ZIP_NAME
        download_file(self._dataset_uri, archive)
        extract_file(archive, scratch)

        splits: Dict[str, Path] = {}
        for name, err in _SPLIT_ERRORS.items():
            d = scratch / name
            if not d.exists():
                raise FileNotFoundError(err.format(d))
            splits[name] = d
        return splits
