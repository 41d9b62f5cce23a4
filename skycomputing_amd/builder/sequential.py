"""Tuple-splatting sequential container.

Capability parity with the reference's SequentialWrapper
(reference: scaelum/builder/sequential_wrapper.py:8-20): a tuple/list output
of layer i is splatted into layer i+1's ``forward(*args)`` — required because
the BERT pipeline layers pass (hidden, mask, ...) tuples.
"""

from __future__ import annotations

import torch.nn as nn


class SequentialWrapper(nn.Sequential):
    def forward(self, *inputs):
        out = inputs
        for module in self:
            if isinstance(out, (tuple, list)):
                out = module(*out)
            else:
                out = module(out)
        return out
