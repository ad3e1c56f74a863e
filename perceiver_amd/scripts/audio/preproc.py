"""MIDI dataset preprocessing CLI (parity: reference scripts/audio/preproc.py)."""
from __future__ import annotations

import argparse

from perceiver_amd.data.audio import GiantMidiPianoDataModule, MaestroV3DataModule

DATASETS = {"giantmidi-piano": GiantMidiPianoDataModule, "maestro-v3": MaestroV3DataModule}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("dataset", choices=sorted(DATASETS))
    p.add_argument("--max_seq_len", type=int, default=2048)
    p.add_argument("--preproc_workers", type=int, default=None)
    args = p.parse_args()

    dm = DATASETS[args.dataset](max_seq_len=args.max_seq_len, preproc_workers=args.preproc_workers)
    dm.prepare_data()
    print(f"preprocessed -> {dm.preproc_dir}")


if __name__ == "__main__":
    main()
