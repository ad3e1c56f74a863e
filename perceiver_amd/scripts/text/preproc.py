"""Dataset preprocessing CLI (parity: reference scripts/text/preproc.py):
tokenize/chunk/cache a text dataset ahead of training.

    python -m perceiver_amd.scripts.text.preproc wikitext --task clm --max_seq_len 4096 ...
"""
from __future__ import annotations

import argparse

from perceiver_amd.data.text import (
    BookCorpusDataModule,
    BookCorpusOpenDataModule,
    Enwik8DataModule,
    ImdbDataModule,
    Task,
    WikipediaDataModule,
    WikiTextDataModule,
)

DATASETS = {
    "wikitext": WikiTextDataModule,
    "enwik8": Enwik8DataModule,
    "imdb": ImdbDataModule,
    "bookcorpus": BookCorpusDataModule,
    "bookcorpusopen": BookCorpusOpenDataModule,
    "wikipedia": WikipediaDataModule,
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("dataset", choices=sorted(DATASETS))
    p.add_argument("--tokenizer", default="deepmind/language-perceiver")
    p.add_argument("--task", default="mlm", choices=[t.name for t in Task])
    p.add_argument("--max_seq_len", type=int, default=2048)
    p.add_argument("--add_special_tokens", action="store_true")
    p.add_argument("--add_eos_token", action="store_true")
    p.add_argument("--static_masking", action="store_true")
    p.add_argument("--preproc_workers", type=int, default=None)
    args = p.parse_args()

    dm = DATASETS[args.dataset](
        tokenizer=args.tokenizer, task=Task[args.task], max_seq_len=args.max_seq_len,
        add_special_tokens=args.add_special_tokens, add_eos_token=args.add_eos_token,
        static_masking=args.static_masking, preproc_workers=args.preproc_workers,
    )
    dm.prepare_data()
    print(f"preprocessed -> {dm.preproc_dir}")


if __name__ == "__main__":
    main()
