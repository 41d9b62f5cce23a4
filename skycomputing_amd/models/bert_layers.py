"""Pipeline-splittable BERT layer zoo.

Capability parity with the reference's registered BERT pieces (reference:
scaelum/model/bert_layers.py:21-395): the encoder block is split into THREE
registry layers (Head = self-attention + self-output, Body = intermediate
FFN, Tail = output FFN + residual LN) so the allocator can cut at a finer
granularity, plus BertEmbeddings / BertPooler / BertTailForClassification.

MI355X-native differences from the reference:
  * every fusable op goes through skycomputing_amd.ops (hand-written gfx950
    HIP kernels on GPU: fused LayerNorm, bias-GELU, masked softmax, fused
    embedding, counter-RNG dropout) instead of eager torch / optional apex;
  * compute dtype is bf16 on GPU (fp32 accumulation inside kernels);
  * each layer reports analytic FLOPs and activation/parameter bytes for the
    allocator's model benchmark (replacing pthflops jit tracing,
    reference: scaelum/dynamics/estimator.py:76-82).

Dataflow tuples between pipeline stages:
  BertEmbeddings(ids, type_ids, attn_mask)  -> (hidden, ext_mask)
  BertLayer_Head(hidden, ext_mask)          -> (attn_out, ext_mask)
  BertLayer_Body(attn_out, ext_mask)        -> (attn_out, intermediate, ext_mask)
  BertLayer_Tail(attn_out, interm, ext_mask)-> (hidden', ext_mask)
  BertPooler(hidden, ext_mask)              -> pooled [B, H]
  BertTailForClassification(pooled)         -> logits [B, C]
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops
from ..registry import LAYER
from .bert import BertConfig


def _cfg(config) -> BertConfig:
    if isinstance(config, BertConfig):
        return config
    if isinstance(config, dict):
        return BertConfig.from_dict(config)
    raise TypeError(f"expected BertConfig or dict, got {type(config)}")


class SkyLayerNorm(nn.Module):
    """LayerNorm with optional fused residual add, dispatching to the HIP
    fused kernel on GPU (replaces apex FusedLayerNorm in the reference,
    scaelum/model/bert_layers.py:128-168)."""

    def __init__(self, hidden_size: int, eps: float = 1e-12):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.bias = nn.Parameter(torch.zeros(hidden_size))
        self.eps = eps

    def forward(self, x, residual=None, dropout_p: float = 0.0, training: bool = False):
        return ops.layer_norm(x, self.weight, self.bias, self.eps, residual,
                              dropout_p, training)


class LinearActivation(nn.Module):
    """Fused linear + bias + activation
    (reference: scaelum/model/bert_layers.py:60-108)."""

    def __init__(self, in_features: int, out_features: int, act: str = "gelu", bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.act = act
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        return ops.linear_act(x, self.weight, self.bias, self.act)


class SkyLinear(nn.Module):
    """nn.Linear drop-in whose backward computes dbias with the HIP column
    reduction (ops.linear); GEMMs stay on hipBLASLt."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        return ops.linear(x, self.weight, self.bias)


class BertSelfAttention(nn.Module):
    """Multi-head self-attention core
    (reference: scaelum/model/bert_layers.py:215-275).

    MI355X design: ONE fused QKV GEMM [B*S,H]x[H,3H] instead of three
    H x H GEMMs (bigger tiles fill the 256 CUs; one launch instead of
    three) — SURVEY.md §2c row 1's fused-QKV option."""

    def __init__(self, config: BertConfig):
        super().__init__()
        if config.hidden_size % config.num_attention_heads != 0:
            raise ValueError("hidden_size must be divisible by num_attention_heads")
        self.num_heads = config.num_attention_heads
        self.head_dim = config.hidden_size // config.num_attention_heads
        H = config.hidden_size
        self.qkv = SkyLinear(H, 3 * H)
        self.dropout_p = config.attention_probs_dropout_prob

    def forward(self, hidden, ext_mask):
        B, S, H = hidden.shape
        qkv = self.qkv(hidden).view(B, S, 3, self.num_heads, self.head_dim)
        return ops.attention(qkv, ext_mask, self.dropout_p, self.training)


class BertSelfOutput(nn.Module):
    """Attention output projection + dropout + residual LN
    (reference: scaelum/model/bert_layers.py:278-289)."""

    def __init__(self, config: BertConfig):
        super().__init__()
        H = config.hidden_size
        self.dense = SkyLinear(H, H)
        self.layer_norm = SkyLayerNorm(H, config.layer_norm_eps)
        self.dropout_p = config.hidden_dropout_prob

    def forward(self, hidden, residual):
        x = self.dense(hidden)
        # dropout fused into the LN kernel: LN(dropout(x) + residual)
        return self.layer_norm(x, residual=residual,
                               dropout_p=self.dropout_p, training=self.training)


@LAYER.register_module
class BertEmbeddings(nn.Module):
    """Word+position+type embeddings, LN, dropout; also builds the additive
    extended attention mask (1-m)*-10000 carried down the pipeline
    (reference: scaelum/model/bert_layers.py:171-212)."""

    def __init__(self, config):
        super().__init__()
        config = _cfg(config)
        self.config = config
        H = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, H)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, H)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, H)
        self.layer_norm = SkyLayerNorm(H, config.layer_norm_eps)
        self.dropout_p = config.hidden_dropout_prob
        for e in (self.word_embeddings, self.position_embeddings, self.token_type_embeddings):
            nn.init.normal_(e.weight, std=config.initializer_range)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        B, S = input_ids.shape
        dev = input_ids.device
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        if attention_mask is None:
            attention_mask = torch.ones(B, S, device=dev, dtype=torch.long)
        # additive extended mask, broadcast to [B,1,1,S]
        ext_mask = (1.0 - attention_mask[:, None, None, :].to(self.word_embeddings.weight.dtype)) * -10000.0
        position_ids = torch.arange(S, device=dev, dtype=torch.long).unsqueeze(0).expand(B, S)
        emb = ops.embedding_fused(
            input_ids, token_type_ids, position_ids,
            self.word_embeddings.weight, self.position_embeddings.weight,
            self.token_type_embeddings.weight,
            self.layer_norm.weight, self.layer_norm.bias, self.layer_norm.eps,
        )
        emb = ops.dropout(emb, self.dropout_p, self.training)
        return emb, ext_mask

    def layer_flops(self, batch: int, seq: int) -> float:
        # gathers+LN are bandwidth ops; count the LN flops only (negligible)
        return 10.0 * batch * seq * self.config.hidden_size

    def activation_numel(self, batch: int, seq: int) -> int:
        return batch * seq * self.config.hidden_size + batch * seq


@LAYER.register_module(name="BertLayer_Head")
class BertLayerHead(nn.Module):
    """Self-attention + self-output third of an encoder block
    (reference: scaelum/model/bert_layers.py:330-339)."""

    def __init__(self, config):
        super().__init__()
        config = _cfg(config)
        self.config = config
        self.attention = BertSelfAttention(config)
        self.output = BertSelfOutput(config)

    def forward(self, hidden, ext_mask):
        attn = self.attention(hidden, ext_mask)
        out = self.output(attn, hidden)
        return out, ext_mask

    def layer_flops(self, batch: int, seq: int) -> float:
        H = self.config.hidden_size
        # QKV + output projections: 4 GEMMs of [B*S,H]x[H,H]; attention 2 batched GEMMs
        gemm = 4 * 2.0 * batch * seq * H * H
        attn = 2 * 2.0 * batch * self.config.num_attention_heads * seq * seq * (
            H // self.config.num_attention_heads
        )
        return gemm + attn

    def activation_numel(self, batch: int, seq: int) -> int:
        H = self.config.hidden_size
        h = self.config.num_attention_heads
        return 6 * batch * seq * H + 2 * batch * h * seq * seq

    def param_count(self) -> int:
        return sum(p.numel() for p in self.parameters())


@LAYER.register_module(name="BertLayer_Body")
class BertLayerBody(nn.Module):
    """Intermediate FFN (fused bias-GELU) third of an encoder block
    (reference: scaelum/model/bert_layers.py:342-351)."""

    def __init__(self, config):
        super().__init__()
        config = _cfg(config)
        self.config = config
        self.intermediate = LinearActivation(
            config.hidden_size, config.intermediate_size, act=config.hidden_act
        )

    def forward(self, attn_out, ext_mask):
        inter = self.intermediate(attn_out)
        return attn_out, inter, ext_mask

    def layer_flops(self, batch: int, seq: int) -> float:
        return 2.0 * batch * seq * self.config.hidden_size * self.config.intermediate_size

    def activation_numel(self, batch: int, seq: int) -> int:
        return batch * seq * (self.config.hidden_size + self.config.intermediate_size)


@LAYER.register_module(name="BertLayer_Tail")
class BertLayerTail(nn.Module):
    """Output FFN + dropout + residual LN third of an encoder block
    (reference: scaelum/model/bert_layers.py:354-363)."""

    def __init__(self, config):
        super().__init__()
        config = _cfg(config)
        self.config = config
        self.dense = SkyLinear(config.intermediate_size, config.hidden_size)
        self.layer_norm = SkyLayerNorm(config.hidden_size, config.layer_norm_eps)
        self.dropout_p = config.hidden_dropout_prob

    def forward(self, attn_out, inter, ext_mask):
        x = self.dense(inter)
        hidden = self.layer_norm(x, residual=attn_out,
                                 dropout_p=self.dropout_p, training=self.training)
        return hidden, ext_mask

    def layer_flops(self, batch: int, seq: int) -> float:
        return 2.0 * batch * seq * self.config.hidden_size * self.config.intermediate_size

    def activation_numel(self, batch: int, seq: int) -> int:
        return batch * seq * (2 * self.config.hidden_size + self.config.intermediate_size)


@LAYER.register_module
class BertPooler(nn.Module):
    """[CLS] slice + dense + tanh (reference: scaelum/model/bert_layers.py:381-395)."""

    def __init__(self, config):
        super().__init__()
        config = _cfg(config)
        self.config = config
        self.dense_act = LinearActivation(config.hidden_size, config.hidden_size, act="tanh")

    def forward(self, hidden, ext_mask=None):
        return self.dense_act(hidden[:, 0])

    def layer_flops(self, batch: int, seq: int) -> float:
        return 2.0 * batch * self.config.hidden_size * self.config.hidden_size

    def activation_numel(self, batch: int, seq: int) -> int:
        return batch * self.config.hidden_size


@LAYER.register_module
class BertTailForClassification(nn.Module):
    """Dropout + classifier head
    (reference: scaelum/model/bert_layers.py:366-378)."""

    def __init__(self, config, num_class: int = 3):
        super().__init__()
        config = _cfg(config)
        self.config = config
        self.num_class = num_class
        self.dropout_p = config.hidden_dropout_prob
        self.classifier = nn.Linear(config.hidden_size, num_class)

    def forward(self, pooled):
        x = ops.dropout(pooled, self.dropout_p, self.training)
        return self.classifier(x)

    def layer_flops(self, batch: int, seq: int) -> float:
        return 2.0 * batch * self.config.hidden_size * self.num_class

    def activation_numel(self, batch: int, seq: int) -> int:
        return batch * self.num_class


def bert_pipeline_config(
    num_encoder_layers: int,
    bert_config: dict | None = None,
    num_class: int = 3,
) -> list[dict]:
    """Build the layer-config list for an N-encoder-layer BERT classifier,
    mirroring the reference experiment's layer list shape
    (reference: experiment/config.py:32-49)."""
    bc = dict(bert_config or {})
    layers: list[dict] = [dict(layer_type="BertEmbeddings", config=bc)]
    for _ in range(num_encoder_layers):
        layers.append(dict(layer_type="BertLayer_Head", config=bc))
        layers.append(dict(layer_type="BertLayer_Body", config=bc))
        layers.append(dict(layer_type="BertLayer_Tail", config=bc))
    layers.append(dict(layer_type="BertPooler", config=bc))
    layers.append(dict(layer_type="BertTailForClassification", config=bc, num_class=num_class))
    return layers
