"""BERT hyperparameter config.

Capability parity with the reference's BertConfig
(reference: scaelum/model/bert.py:6-99): construct from kwargs, dict or a
json file; used by every registered BERT pipeline layer.
"""

from __future__ import annotations

import json


class BertConfig:
    def __init__(
        self,
        vocab_size_or_config_json_file=30522,
        hidden_size: int = 1024,
        num_hidden_layers: int = 24,
        num_attention_heads: int = 16,
        intermediate_size: int = 4096,
        hidden_act: str = "gelu",
        hidden_dropout_prob: float = 0.1,
        attention_probs_dropout_prob: float = 0.1,
        max_position_embeddings: int = 512,
        type_vocab_size: int = 2,
        initializer_range: float = 0.02,
        layer_norm_eps: float = 1e-12,
        output_all_encoded_layers: bool = False,
    ):
        if isinstance(vocab_size_or_config_json_file, str):
            with open(vocab_size_or_config_json_file) as f:
                data = json.load(f)
            for k, v in data.items():
                setattr(self, k, v)
            return
        self.vocab_size = int(vocab_size_or_config_json_file)
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.output_all_encoded_layers = output_all_encoded_layers

    @classmethod
    def from_dict(cls, d: dict) -> "BertConfig":
        cfg = cls(d.get("vocab_size", 30522))
        for k, v in d.items():
            setattr(cfg, k, v)
        return cfg

    @classmethod
    def from_json_file(cls, path: str) -> "BertConfig":
        return cls(path)

    def to_dict(self) -> dict:
        return dict(self.__dict__)

    def __repr__(self):
        return f"BertConfig({json.dumps(self.to_dict(), indent=2)})"
