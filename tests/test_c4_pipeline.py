"""C4 streaming pipeline (reference data/text/c4.py): tokenize -> EOS-joined
token stream -> fixed/random chunking -> pad + shift-by-one collation, plus
rank sharding via split_dataset_by_node — all exercised offline on a synthetic
iterable dataset."""
import datasets
import pytest
import torch
from transformers import PerceiverTokenizer

from perceiver_amd.data.text.c4 import C4DataModule


def _synthetic_stream(n=40):
    return datasets.Dataset.from_dict({
        "text": [f"document number {i} with some words " * (3 + i % 5) for i in range(n)],
        "timestamp": ["2020-01-01" for _ in range(n)],
        "url": [f"http://x/{i}" for i in range(n)],
    }).to_iterable_dataset()


def _dm(**kw):
    base = dict(tokenizer=PerceiverTokenizer(), max_seq_len=64, batch_size=4,
                num_train_workers=0, num_valid_workers=0, pin_memory=False,
                rank=0, world_size=1)
    base.update(kw)
    return C4DataModule(**base)


def test_c4_pipeline_chunks_and_shifts():
    dm = _dm()
    ds = dm._create_pipeline(_synthetic_stream())
    # Collator.__call__ yields the (labels, input_ids, pad_mask) batch contract
    y, x, pad = next(iter(torch.utils.data.DataLoader(ds, collate_fn=dm.collator, batch_size=4)))
    assert x.shape == (4, 64) and y.shape == (4, 64) and pad.shape == (4, 64)
    # shift-by-one: labels are the next-token stream
    assert torch.equal(x[:, 1:], y[:, :-1])
    assert not pad.any()  # fixed-length chunks need no padding


def test_c4_random_chunk_lengths_within_bounds():
    dm = _dm(min_seq_len=16)
    ds = dm._create_pipeline(_synthetic_stream(), min_seq_len=16)
    lens = [len(ex["input_ids"]) for ex, _ in zip(ds, range(20))]
    assert all(17 <= l <= 65 for l in lens), sorted(set(lens))
    assert len(set(lens)) > 1  # actually random


def test_c4_rank_sharding_is_disjoint():
    from datasets.distributed import split_dataset_by_node

    full = [ex["text"] for ex in _synthetic_stream()]
    shard0 = [ex["text"] for ex in split_dataset_by_node(_synthetic_stream(), rank=0, world_size=2)]
    shard1 = [ex["text"] for ex in split_dataset_by_node(_synthetic_stream(), rank=1, world_size=2)]
    assert not (set(shard0) & set(shard1))
    assert sorted(shard0 + shard1) == sorted(full)
