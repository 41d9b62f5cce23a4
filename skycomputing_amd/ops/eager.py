"""Eager (plain PyTorch) implementations of every fused op.

These serve two purposes:
  * the CPU execution path (tests, gloo multi-process plumbing), and
  * the fp32 numerics reference that every HIP kernel is tested against
    (tests/test_ops_gpu.py compares HIP bf16 kernels vs these in fp32).

The op set mirrors the reference's BERT hot path (reference:
scaelum/model/bert_layers.py:21-168 — gelu/bias_gelu/swish/bias_tanh,
LayerNorm, LinearActivation; :249-275 masked softmax) but is organized as a
fusion-shaped functional API so the HIP kernels slot in 1:1.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F


def gelu(x: torch.Tensor) -> torch.Tensor:
    """erf-formula GELU (the reference's formula, not tanh-approx)."""
    return x * 0.5 * (1.0 + torch.erf(x / math.sqrt(2.0)))


def bias_gelu(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    return gelu(x + bias)


def swish(x: torch.Tensor) -> torch.Tensor:
    return x * torch.sigmoid(x)


def bias_tanh(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    return torch.tanh(x + bias)


ACT_FNS = {"gelu": gelu, "relu": F.relu, "swish": swish, "tanh": torch.tanh}


def layer_norm(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: torch.Tensor,
    eps: float = 1e-12,
    residual: torch.Tensor | None = None,
) -> torch.Tensor:
    """LayerNorm over the last dim with optional fused residual add.

    Accumulates in fp32 regardless of input dtype (what the HIP kernel does).
    """
    if residual is not None:
        x = x + residual
    orig_dtype = x.dtype
    y = F.layer_norm(x.float(), (x.shape[-1],), weight.float(), bias.float(), eps)
    return y.to(orig_dtype)


def masked_softmax(scores: torch.Tensor, mask: torch.Tensor | None) -> torch.Tensor:
    """softmax(scores + mask) over the last dim, fp32 accumulation.

    ``mask`` is the additive extended attention mask ((1-m) * -10000,
    broadcastable to ``scores``), as built by the embeddings layer.
    """
    orig_dtype = scores.dtype
    s = scores.float()
    if mask is not None:
        s = s + mask.float()
    return F.softmax(s, dim=-1).to(orig_dtype)


def attention_context(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    mask: torch.Tensor | None,
    dropout_p: float = 0.0,
    training: bool = False,
) -> torch.Tensor:
    """Full attention: softmax(QK^T/sqrt(d) + mask) V.

    q/k/v: [B, h, S, d]. Returns [B, h, S, d].
    """
    d = q.shape[-1]
    scores = torch.matmul(q, k.transpose(-1, -2)) / math.sqrt(d)
    probs = masked_softmax(scores, mask)
    if dropout_p > 0.0 and training:
        probs = F.dropout(probs, p=dropout_p, training=True)
    return torch.matmul(probs, v)


def linear_act(
    x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor | None, act: str = "gelu"
) -> torch.Tensor:
    """Linear + bias + activation in one op (GEMM-epilogue fusion target).

    Mirrors the reference's LinearActivation fused module
    (reference: scaelum/model/bert_layers.py:60-108)."""
    y = F.linear(x, weight)
    if bias is not None:
        if act == "gelu":
            return bias_gelu(y, bias)
        if act == "tanh":
            return bias_tanh(y, bias)
        y = y + bias
    return ACT_FNS[act](y) if act != "none" else y


def embedding_fused(
    input_ids: torch.Tensor,
    token_type_ids: torch.Tensor,
    position_ids: torch.Tensor,
    word_emb: torch.Tensor,
    pos_emb: torch.Tensor,
    type_emb: torch.Tensor,
    ln_weight: torch.Tensor,
    ln_bias: torch.Tensor,
    eps: float = 1e-12,
) -> torch.Tensor:
    """3-way embedding gather + add + LayerNorm (dropout applied by caller)."""
    e = (
        F.embedding(input_ids, word_emb)
        + F.embedding(position_ids, pos_emb)
        + F.embedding(token_type_ids, type_emb)
    )
    return layer_norm(e, ln_weight, ln_bias, eps)


@torch.no_grad()
def sgd_step(
    params: list[torch.Tensor],
    grads: list[torch.Tensor],
    lr: float,
    momentum: float = 0.0,
    weight_decay: float = 0.0,
    momentum_bufs: list[torch.Tensor] | None = None,
    master_params: list[torch.Tensor] | None = None,
):
    """Multi-tensor SGD. With ``master_params`` (fp32), the update is applied
    to the master copy and the (possibly bf16) param is refreshed from it."""
    for i, (p, g) in enumerate(zip(params, grads)):
        if g is None:
            continue
        target = master_params[i] if master_params is not None else p
        # the update is computed in fp32 (momentum buffers are fp32 even
        # when the params are bf16 without masters — see FusedSGD)
        gf = g.float()
        if weight_decay != 0.0:
            gf = gf.add(target.float(), alpha=weight_decay)
        if momentum != 0.0:
            buf = momentum_bufs[i]
            buf.mul_(momentum).add_(gf)
            gf = buf
        if target.dtype == torch.float32:
            target.add_(gf, alpha=-lr)
        else:
            target.copy_((target.float().add_(gf, alpha=-lr)).to(target.dtype))
        if master_params is not None:
            p.copy_(target.to(p.dtype))
