"""Benchmark input factories.

Capability parity with the reference generators
(reference: scaelum/dataset/data_generator.py:10-34), with the
DataloaderGenerator iterator bug fixed (the reference re-built the iterator
every call and always returned the first batch, data_generator.py:30-34).
"""

from __future__ import annotations

import torch

from ..registry import DATA_GENERATOR


class BaseGenerator:
    def generate(self):
        raise NotImplementedError


@DATA_GENERATOR.register_module
class RandomTensorGenerator(BaseGenerator):
    def __init__(self, seed: int = 0, **tensor_kwargs):
        self.kwargs = tensor_kwargs
        self.g = torch.Generator().manual_seed(seed)

    def generate(self):
        size = self.kwargs.get("size")
        return torch.rand(*size, generator=self.g)


@DATA_GENERATOR.register_module
class RandomTokenGenerator(BaseGenerator):
    """MNLI-shaped probe batch: (input_ids, attention_mask, token_type_ids)."""

    def __init__(self, batch_size: int = 32, seq_len: int = 128, vocab_size: int = 30522, seed: int = 0):
        self.batch_size, self.seq_len, self.vocab_size = batch_size, seq_len, vocab_size
        self.g = torch.Generator().manual_seed(seed)

    def generate(self):
        ids = torch.randint(0, self.vocab_size, (self.batch_size, self.seq_len), generator=self.g)
        mask = torch.ones(self.batch_size, self.seq_len, dtype=torch.long)
        type_ids = torch.zeros(self.batch_size, self.seq_len, dtype=torch.long)
        return ids, mask, type_ids


@DATA_GENERATOR.register_module
class DataloaderGenerator(BaseGenerator):
    def __init__(self, dataloader_cfg: dict):
        from ..builder import build_dataloader_from_cfg

        self.dataloader = build_dataloader_from_cfg(**dataloader_cfg)
        self._iter = iter(self.dataloader)

    def generate(self):
        try:
            batch = next(self._iter)
        except StopIteration:
            self._iter = iter(self.dataloader)
            batch = next(self._iter)
        data, _label = batch
        return data
