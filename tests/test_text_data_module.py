"""Text data pipeline behavioral tests (contract of the reference's
tests/text_data_module_test.py, SURVEY.md §4 category 6), fully offline: a local
DatasetDict + the byte-level PerceiverTokenizer (constructs without network)."""
import numpy as np
import pytest
import torch
from transformers import PerceiverTokenizer

from perceiver_amd.data.text.common import CLMDataset, RandomShiftDataset, Task, TextDataModule

TEXTS = [
    "The quick brown fox jumps over the lazy dog. " * 8,
    "Perceiver models scale attention through a latent bottleneck. " * 6,
    "MI355X has two hundred and fifty six compute units. " * 7,
    "Short text.",
] * 12


class LocalTextDataModule(TextDataModule):
    """TextDataModule over an in-memory corpus."""

    def load_source_dataset(self):
        from datasets import Dataset, DatasetDict

        ds = Dataset.from_dict({"text": TEXTS})
        return DatasetDict(train=ds, valid=Dataset.from_dict({"text": TEXTS[:8]}))


def make_dm(tmp_path, task=Task.mlm, max_seq_len=64, **kw):
    return LocalTextDataModule(
        dataset_dir=str(tmp_path), tokenizer=PerceiverTokenizer(), max_seq_len=max_seq_len,
        task=task, num_workers=0, preproc_workers=1, batch_size=4, pin_memory=False, **kw,
    )


@pytest.fixture(scope="module")
def mlm_dm(tmp_path_factory):
    dm = make_dm(tmp_path_factory.mktemp("mlm"), task=Task.mlm)
    dm.prepare_data()
    dm.setup()
    return dm


def test_mlm_chunking(mlm_dm):
    ex = mlm_dm.ds_train[0]
    assert len(ex["input_ids"]) == 64
    assert len(ex["word_ids"]) == 64


def test_mlm_batch_contract_and_masking_rate(mlm_dm):
    np.random.seed(0)
    loader = mlm_dm.train_dataloader()
    masked_frac = []
    mask_token_share = []
    for labels, input_ids, pad_mask in loader:
        assert labels.shape == input_ids.shape == pad_mask.shape
        assert pad_mask.dtype == torch.bool
        sel = labels != -100
        masked_frac.append(sel.float().mean().item())
        if sel.any():
            mask_token_share.append(
                (input_ids[sel] == mlm_dm.tokenizer.mask_token_id).float().mean().item()
            )
    # ~mask_prob positions masked; ~80% of them replaced by the mask token
    assert 0.05 < np.mean(masked_frac) < 0.30
    assert 0.6 < np.mean(mask_token_share) < 0.95


def test_mlm_dynamic_masking_differs_between_epochs(mlm_dm):
    np.random.seed(1)
    l1, x1, _ = next(iter(mlm_dm.train_dataloader()))
    np.random.seed(2)
    l2, x2, _ = next(iter(mlm_dm.train_dataloader()))
    # dynamic: different draws mask different positions (shuffled loaders; just
    # require *some* difference across the epoch batches)
    assert not (torch.equal(l1, l2) and torch.equal(x1, x2))


def test_static_masking_is_deterministic(tmp_path):
    dm = make_dm(tmp_path, task=Task.mlm, static_masking=True)
    dm.prepare_data()
    dm.setup()
    ex1 = dm.ds_train[0]
    ex2 = dm.ds_train[0]
    assert ex1["input_ids"] == ex2["input_ids"]
    assert ex1["labels"] == ex2["labels"]
    # collator for static masking is the default (no re-masking)
    from perceiver_amd.data.text.collator import DefaultCollator

    assert isinstance(dm.collator, DefaultCollator)


def test_clm_shift_property(tmp_path):
    dm = make_dm(tmp_path, task=Task.clm)
    dm.prepare_data()
    dm.setup()
    assert isinstance(dm.ds_train, CLMDataset)
    loader = dm.train_dataloader()
    labels, x, pad = next(iter(loader))
    assert x.shape[1] == 64
    # label[i] == input[i+1] (shift-by-one over the +1-sized chunks)
    assert torch.equal(x[:, 1:], labels[:, :-1])


def test_clf_keeps_labels(tmp_path):
    from datasets import Dataset, DatasetDict

    class ClfDM(TextDataModule):
        def load_source_dataset(self):
            ds = Dataset.from_dict({"text": TEXTS, "label": [i % 2 for i in range(len(TEXTS))]})
            return DatasetDict(train=ds, valid=ds)

    dm = ClfDM(dataset_dir=str(tmp_path), tokenizer=PerceiverTokenizer(), max_seq_len=64,
               task=Task.clf, num_workers=0, preproc_workers=1, batch_size=4, pin_memory=False)
    dm.prepare_data()
    dm.setup()
    y, x, pad = next(iter(dm.train_dataloader()))
    assert y.shape == (4,)
    assert set(y.tolist()) <= {0, 1}


def test_random_truncation(tmp_path):
    torch.manual_seed(5)
    dm = make_dm(tmp_path, task=Task.clm, random_train_truncation=True, random_min_seq_len=16)
    dm.prepare_data()
    dm.setup()
    lengths = {next(iter(dm.train_dataloader()))[1].shape[1] for _ in range(8)}
    assert all(16 <= n <= 64 for n in lengths)
    assert len(lengths) > 1  # actually random


def test_random_shift_dataset():
    base = [{"input_ids": list(range(i * 10, i * 10 + 10))} for i in range(3)]

    class L(torch.utils.data.Dataset):
        def __getitem__(self, i):
            return base[i]

        def __len__(self):
            return len(base)

    torch.manual_seed(0)
    ds = RandomShiftDataset(L())
    assert len(ds) == 2
    ex = ds[0]
    assert len(ex["input_ids"]) == 10
    # suffix of record 0 + prefix of record 1
    ids = ex["input_ids"]
    split = next((i for i in range(1, 10) if ids[i] < ids[i - 1]), 10)
    assert ids[:split] == list(range(10 - split, 10)) or split == 10


def test_left_padding_side(tmp_path):
    dm = make_dm(tmp_path, task=Task.clm, padding_side="left")
    assert dm.tokenizer.padding_side == "left"


def test_preproc_dir_hash_changes_with_config(tmp_path):
    dm1 = make_dm(tmp_path, task=Task.mlm)
    dm2 = make_dm(tmp_path, task=Task.clm)
    dm3 = make_dm(tmp_path, task=Task.mlm, max_seq_len=128)
    assert len({dm1.preproc_dir, dm2.preproc_dir, dm3.preproc_dir}) == 3
