"""Download/extract helpers for audio datasets (parity: reference data/audio/utils.py)."""
from __future__ import annotations

import shutil
import urllib.request
import zipfile
from pathlib import Path


def download_file(uri: str, target_file: Path) -> None:
    with urllib.request.urlopen(uri) as response, open(target_file, "wb") as f:
        shutil.copyfileobj(response, f)


def extract_file(archive_file: Path, target_dir: Path) -> None:
    with zipfile.ZipFile(archive_file, "r") as z:
        z.extractall(target_dir)
