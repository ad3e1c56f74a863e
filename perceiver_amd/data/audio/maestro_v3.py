"""Maestro V3 symbolic-audio data module.

The Maestro archive ships every split in one zip plus a metadata JSON that
assigns each MIDI file to train/validation/test. This module downloads the
archive once, sorts the train and validation files into the split-directory
layout `SymbolicAudioDataModule` expects, and drops the test split (the
framework validates on the validation split only, matching the reference
behavior at data/audio/maestro_v3.py).
"""
from __future__ import annotations

import json
import os
import shutil
from pathlib import Path
from typing import Any, Dict

from perceiver_amd.data.audio.symbolic import SymbolicAudioDataModule
from perceiver_amd.data.audio.utils import download_file, extract_file

_ZIP_NAME = "maestro-v3.0.0-midi.zip"
_ARCHIVE_ROOT = "maestro-v3.0.0"
_DEFAULT_URI = f"https://martin-krasser.com/perceiver/data/midi/{_ZIP_NAME}"


def _read_split_assignments(archive_dir: Path) -> Dict[str, str]:
    """midi path -> split name, from the archive's metadata JSON."""
    meta_path = archive_dir / f"{_ARCHIVE_ROOT}.json"
    if not meta_path.exists():
        raise FileNotFoundError(f"Could not find Maestro v3 dataset meta file (expected=`{meta_path}`)")
    meta = json.loads(meta_path.read_text())
    filenames = meta["midi_filename"]
    split_of = meta["split"]
    return {filenames[key]: split_of[key] for key in filenames}


class MaestroV3DataModule(SymbolicAudioDataModule):
    """Symbolic-audio training on the Maestro v3 piano-performance corpus."""

    def __init__(
        self,
        *args: Any,
        dataset_uri: str = _DEFAULT_URI,
        dataset_dir: str = os.path.join(".cache", "maestro-v3-midi"),
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
        targets = {name: root / "_splits" / name for name in ("train", "valid")}
        for d in targets.values():
            d.mkdir(parents=True)

        archive = scratch / _ZIP_NAME
        download_file(self._dataset_uri, archive)
        extract_file(archive, scratch)

        extracted = scratch / _ARCHIVE_ROOT
        if not extracted.exists():
            raise FileNotFoundError(
                f"Could not find Maestro v3 dataset directory in downloaded dataset (expected=`{extracted}`)"
            )

        for rel_path, split in _read_split_assignments(extracted).items():
            if split == "test":
                continue
            dest = targets["train" if split == "train" else "valid"] / rel_path
            dest.parent.mkdir(parents=True, exist_ok=True)
            shutil.move(extracted / rel_path, dest)

        shutil.rmtree(scratch)
        return dict(targets)
