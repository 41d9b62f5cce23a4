"""Datasets.

Capability parity with the reference dataset zoo (reference:
scaelum/dataset/dataset.py:14-48 RandomMlpDataset/CIFAR10Dataset and
scaelum/dataset/bert_dataset.py:16-94 GlueDataset). Because this image has
no network access, GlueDataset tokenizes from an on-disk GLUE directory
when one exists (via the `transformers` tokenizer if a vocab is present)
and SyntheticGlueDataset provides the MNLI-*shaped* deterministic synthetic
workload used by benchmarks (BASELINE.json: synthetic data, random-init
weights). Sample schema matches the reference:
``((input_ids, attention_mask, token_type_ids), label)``.
"""

from __future__ import annotations

import os

import torch
from torch.utils.data import Dataset

from ..registry import DATASET


@DATASET.register_module
class RandomMlpDataset(Dataset):
    """(reference: scaelum/dataset/dataset.py:14-30)"""

    def __init__(self, size: int = 1024, dim: int = 256, num_class: int = 10, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randn(size, dim, generator=g)
        self.labels = torch.randint(0, num_class, (size,), generator=g)

    def __len__(self):
        return self.data.shape[0]

    def __getitem__(self, i):
        return self.data[i], self.labels[i]


@DATASET.register_module
class CIFAR10Dataset(Dataset):
    """CIFAR10 wrapper (reference: scaelum/dataset/dataset.py:33-48).
    Falls back to a synthetic image tensor set when torchvision/data are
    unavailable (this image has no torchvision and no network)."""

    def __init__(self, root: str = "./data", train: bool = True, size: int = 2048, seed: int = 0):
        self._tv = None
        try:
            from torchvision import datasets as tvd  # pragma: no cover

            if os.path.isdir(root):
                self._tv = tvd.CIFAR10(root=root, train=train, download=False)
        except Exception:
            self._tv = None
        if self._tv is None:
            g = torch.Generator().manual_seed(seed)
            self.data = torch.randn(size, 3, 32, 32, generator=g)
            self.labels = torch.randint(0, 10, (size,), generator=g)

    def __len__(self):
        return len(self._tv) if self._tv is not None else self.data.shape[0]

    def __getitem__(self, i):
        if self._tv is not None:  # pragma: no cover
            img, label = self._tv[i]
            import numpy as np

            return torch.from_numpy(np.array(img)).permute(2, 0, 1).float() / 255.0, label
        return self.data[i], self.labels[i]


@DATASET.register_module
class SyntheticGlueDataset(Dataset):
    """Deterministic MNLI-shaped synthetic data: token ids in [0, vocab),
    full attention mask with a random padded tail, 2-segment type ids,
    3-class labels. Schema identical to the tokenized GlueDataset."""

    def __init__(
        self,
        size: int = 4096,
        max_seq_length: int = 128,
        vocab_size: int = 30522,
        num_class: int = 3,
        seed: int = 0,
    ):
        g = torch.Generator().manual_seed(seed)
        self.input_ids = torch.randint(0, vocab_size, (size, max_seq_length), generator=g)
        lengths = torch.randint(max_seq_length // 2, max_seq_length + 1, (size,), generator=g)
        ar = torch.arange(max_seq_length)[None, :]
        self.attention_mask = (ar < lengths[:, None]).long()
        seg = torch.randint(max_seq_length // 4, 3 * max_seq_length // 4, (size,), generator=g)
        self.token_type_ids = (ar >= seg[:, None]).long() * self.attention_mask
        self.labels = torch.randint(0, num_class, (size,), generator=g)

    def __len__(self):
        return self.input_ids.shape[0]

    def __getitem__(self, i):
        return (
            (self.input_ids[i], self.attention_mask[i], self.token_type_ids[i]),
            self.labels[i],
        )


@DATASET.register_module
class GlueDataset(Dataset):
    """GLUE task dataset (reference: scaelum/dataset/bert_dataset.py:16-94).

    Tokenizes from ``data_dir`` (tsv files + vocab) when present using the
    installed `transformers` BertTokenizer; with no data on disk it refuses
    (use SyntheticGlueDataset for the offline benchmark workload).
    """

    TASK_LABELS = {
        "mnli": ["contradiction", "entailment", "neutral"],
        "mrpc": ["0", "1"],
        "cola": ["0", "1"],
        "sst-2": ["0", "1"],
    }

    def __init__(self, data_dir: str, task: str = "mnli", vocab_file: str | None = None,
                 max_seq_length: int = 128, split: str = "train",
                 cache: bool = True):
        task = task.lower()
        if task not in self.TASK_LABELS:
            raise ValueError(f"unknown GLUE task {task}")
        if not os.path.isdir(data_dir):
            raise FileNotFoundError(
                f"GLUE data dir {data_dir} not found; this image has no network — "
                "use SyntheticGlueDataset for offline runs"
            )
        # tokenized-feature cache next to the data (the reference pickled
        # features too, scaelum/dataset/bert_dataset.py:42-66)
        cache_path = os.path.join(
            data_dir, f"cached_{task}_{split}_{max_seq_length}.pt"
        )
        if cache and os.path.isfile(cache_path):
            import torch as _t

            blob = _t.load(cache_path, weights_only=True)
            self.input_ids = blob["input_ids"]
            self.attention_mask = blob["attention_mask"]
            self.token_type_ids = blob["token_type_ids"]
            self.labels = blob["labels"]
            return
        from transformers import BertTokenizerFast

        vocab = vocab_file or os.path.join(data_dir, "vocab.txt")
        tok = BertTokenizerFast(vocab_file=vocab, do_lower_case=True)
        examples = self._read_examples(data_dir, task, split)
        label_map = {l: i for i, l in enumerate(self.TASK_LABELS[task])}
        texts = [a for a, _, _ in examples]
        pairs = [b for _, b, _ in examples]
        # single-sentence tasks (sst-2, cola) must pass text_pair=None, not
        # a list of empty strings
        pair_arg = pairs if any(pairs) else None
        enc = tok(
            texts, pair_arg,
            padding="max_length", truncation=True, max_length=max_seq_length,
            return_tensors="pt",
        )
        self.input_ids = enc["input_ids"]
        self.attention_mask = enc["attention_mask"]
        self.token_type_ids = enc.get(
            "token_type_ids", torch.zeros_like(self.input_ids)
        )
        self.labels = torch.tensor([label_map[l] for _, _, l in examples], dtype=torch.long)
        if cache:
            try:
                torch.save(
                    dict(input_ids=self.input_ids,
                         attention_mask=self.attention_mask,
                         token_type_ids=self.token_type_ids,
                         labels=self.labels),
                    cache_path,
                )
            except OSError:
                pass  # read-only data dir: skip caching

    @staticmethod
    def _read_examples(data_dir: str, task: str, split: str):
        import csv

        fname = {"train": "train.tsv", "dev": "dev.tsv"}.get(split, f"{split}.tsv")
        path = os.path.join(data_dir, fname)
        rows = []
        with open(path, encoding="utf-8") as f:
            reader = csv.reader(f, delimiter="\t", quotechar=None)
            # CoLA tsvs have NO header row (reference ColaProcessor reads
            # from line 0, scaelum/dataset/glue/processor.py); the other
            # tasks carry one.
            header = next(reader) if task != "cola" else None
            if task == "mnli":
                idx = (header.index("sentence1"), header.index("sentence2"), header.index("gold_label"))
                for r in reader:
                    rows.append((r[idx[0]], r[idx[1]], r[idx[2]]))
            elif task == "mrpc":
                for r in reader:
                    rows.append((r[3], r[4], r[0]))
            elif task in ("cola",):
                for r in reader:
                    rows.append((r[3], "", r[1]))
            elif task in ("sst-2",):
                for r in reader:
                    rows.append((r[0], "", r[1]))
        return rows

    def __len__(self):
        return self.input_ids.shape[0]

    def __getitem__(self, i):
        return (
            (self.input_ids[i], self.attention_mask[i], self.token_type_ids[i]),
            self.labels[i],
        )
