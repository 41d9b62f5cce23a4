"""Public functional op API with device dispatch.

CPU tensors (tests, gloo plumbing) run the eager fp32-reference path; CUDA
tensors run the hand-written gfx950 HIP kernels. On a GPU box with the
extension missing the core ops FAIL LOUDLY (hiplib.require) instead of
silently falling back to eager — set SKY_ALLOW_EAGER_GPU=1 to override for
debugging only.

Op inventory maps 1:1 onto the reference's BERT hot path
(SURVEY.md §2c table); plain unfused GEMMs go through torch (hipBLASLt on
ROCm), fused GEMM+epilogue paths are progressively replaced by MFMA kernels.
"""

from __future__ import annotations

import math
import os

import torch

from . import eager, hiplib
from .eager import ACT_FNS, gelu, swish  # re-export

__all__ = [
    "layer_norm", "bias_gelu", "bias_tanh", "masked_softmax", "dropout",
    "attention_context", "linear_act", "embedding_fused", "sgd_step",
    "gelu", "swish", "ACT_FNS", "hip_available",
]


def hip_available() -> bool:
    return hiplib.available()


def _use_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if hiplib.available():
        return True
    if os.environ.get("SKY_ALLOW_EAGER_GPU") == "1":
        return False
    hiplib.require()  # raises with a build hint
    return False


def layer_norm(x, weight, bias, eps: float = 1e-12, residual=None,
               dropout_p: float = 0.0, training: bool = False):
    """LayerNorm with optional fused residual add and fused PRE-add dropout
    on x: LN(dropout(x) + residual)."""
    p = dropout_p if training else 0.0
    if _use_hip(x):
        from .functions import LayerNormFn

        if p > 0.0 and x.shape[-1] % 8 != 0:
            x = dropout(x, p, True)
            p = 0.0
        return LayerNormFn.apply(x, weight, bias, eps, residual, p)
    if p > 0.0:
        x = torch.nn.functional.dropout(x, p=p, training=True)
    return eager.layer_norm(x, weight, bias, eps, residual)


def bias_gelu(x, bias):
    if _use_hip(x):
        from .functions import BiasGeluFn

        return BiasGeluFn.apply(x, bias)
    return eager.bias_gelu(x, bias)


def bias_tanh(x, bias):
    # tiny (pooler-sized) op: torch native on both devices
    return eager.bias_tanh(x, bias)


def masked_softmax(scores, mask, scale: float = 1.0):
    """softmax(scores*scale + mask) over the last dim of [B,h,Sq,Sk]."""
    if _use_hip(scores):
        from .functions import MaskedSoftmaxFn

        return MaskedSoftmaxFn.apply(scores, mask, scale)
    return eager.masked_softmax(scores * scale if scale != 1.0 else scores, mask)


def dropout(x, p: float, training: bool = True):
    if p <= 0.0 or not training:
        return x
    if _use_hip(x):
        from .functions import DropoutFn

        return DropoutFn.apply(x, p)
    return torch.nn.functional.dropout(x, p=p, training=True)


def attention_context(q, k, v, mask, dropout_p: float = 0.0, training: bool = False):
    """softmax(QK^T/sqrt(d) + mask) V for q/k/v [B,h,S,d].

    GPU path: batched GEMMs via torch (hipBLASLt) + fused HIP masked-softmax
    + seed-regenerated dropout. (A flash-style fused kernel replaces this
    path for long sequences; at the reference workload S=128 the score
    matrix is L2-resident and the batched-GEMM path is the right shape.)
    """
    d = q.shape[-1]
    scale = 1.0 / math.sqrt(d)
    if _use_hip(q):
        scores = torch.matmul(q, k.transpose(-1, -2))
        probs = masked_softmax(scores, mask, scale)
        probs = dropout(probs, dropout_p, training)
        return torch.matmul(probs, v)
    return eager.attention_context(q, k, v, mask, dropout_p, training)


def attention(qkv, mask, dropout_p: float = 0.0, training: bool = False):
    """Multi-head attention over a packed qkv tensor [B, S, 3, h, d];
    returns [B, S, h*d].

    GPU fast path (bf16, d=64, S<=128): ONE fused MFMA kernel
    (FusedAttentionFn). Otherwise: strided-view batched GEMMs + fused
    masked softmax (HIP on GPU, eager on CPU)."""
    B, S, three, h, d = qkv.shape
    scale = 1.0 / math.sqrt(d)
    if (
        qkv.is_cuda and qkv.dtype == torch.bfloat16 and d == 64
        and hiplib.available() and os.environ.get("SKY_NO_FUSED_ATTN") != "1"
    ):
        from .functions import FlashAttentionFn, FusedAttentionFn

        if S <= 128:
            out = FusedAttentionFn.apply(qkv, mask, scale, dropout_p, training)
            return out.reshape(B, S, h * d)
        if S <= 4096 and os.environ.get("SKY_FLASH_ATTN") == "1":
            # our experimental flash-style forward (online softmax, no
            # S x S tensor in forward; backward decomposes). Kept
            # env-selectable; the sdpa path below beats it at every
            # measured S (tools/flash_bench.py), so it is not the default.
            out = FlashAttentionFn.apply(qkv, mask, scale, dropout_p, training)
            return out.reshape(B, S, h * d)
        # long-sequence default: torch-ROCm's flash scaled_dot_product
        # _attention (fwd AND bwd tiled — no S x S tensor in either
        # direction, unlike the decomposed path whose backward
        # materializes scores). Same standing as hipBLASLt for plain
        # GEMMs: library kernels where they win, hand kernels where ours
        # do (S<=128 is the fused MFMA kernel above). Measured at S=512:
        # sdpa ~26 us core vs 108 us decomposed / 176 us flash-v1.
        if os.environ.get("SKY_NO_SDPA") != "1":
            q = qkv[:, :, 0].permute(0, 2, 1, 3)
            k = qkv[:, :, 1].permute(0, 2, 1, 3)
            v = qkv[:, :, 2].permute(0, 2, 1, 3)
            am = mask
            if am is not None:
                am = am.to(qkv.dtype)  # additive [B,1,1,S], broadcast
            ctx = torch.nn.functional.scaled_dot_product_attention(
                q, k, v, attn_mask=am,
                dropout_p=dropout_p if training else 0.0,
            )
            return ctx.permute(0, 2, 1, 3).reshape(B, S, h * d)
    q = qkv[:, :, 0].permute(0, 2, 1, 3)
    k = qkv[:, :, 1].permute(0, 2, 1, 3)
    v = qkv[:, :, 2].permute(0, 2, 1, 3)
    ctx = attention_context(q, k, v, mask, dropout_p, training)
    return ctx.permute(0, 2, 1, 3).reshape(B, S, h * d)


def linear(x, weight, bias=None):
    """Linear with HIP-colsum dbias backward on GPU (GEMMs via hipBLASLt)."""
    if _use_hip(x):
        from .functions import LinearBiasFn

        return LinearBiasFn.apply(x, weight, bias)
    return torch.nn.functional.linear(x, weight, bias)


def linear_act(x, weight, bias, act: str = "gelu"):
    """Linear + bias + activation (the reference's LinearActivation).

    GPU: ONE hipBLASLt GEMM with the GELU_AUX_BIAS epilogue (bias + gelu +
    pre-activation save fused into the GEMM); SKY_NO_HBLT=1 falls back to
    GEMM + the standalone HIP bias+gelu kernel."""
    if _use_hip(x) and act == "gelu" and bias is not None:
        from .functions import LinearGeluFn, _use_hblt

        # GELU_AUX_BIAS epilogue: this hipBLASLt build only supports it for
        # fp32 D (bf16 returns HIPBLAS_STATUS_NOT_SUPPORTED — probed on HW,
        # tools/hblt_probe.py), so the fused linear+gelu is fp32-only and
        # opt-in; the bf16 hot path keeps GEMM + the HIP bias+gelu kernel.
        if (_use_hblt() and os.environ.get("SKY_HBLT_GELU") == "1"
                and x.dtype == torch.float32 and weight.shape[0] % 8 == 0):
            return LinearGeluFn.apply(x, weight, bias)
        # v2 hand-GEMM path: one 256^2 8-phase kernel with the bias+gelu
        # epilogue fused (and the pre-activation stored for backward)
        if x.dtype == torch.bfloat16:
            from .functions import _g2_enabled, _g2_fit

            rows = x.numel() // x.shape[-1]
            if (_g2_enabled("fwd", rows, weight.shape[0], weight.shape[1])
                    and _g2_fit(rows, weight.shape[0], weight.shape[1])):
                return LinearGeluFn.apply(x, weight, bias)
        y = torch.nn.functional.linear(x, weight)
        return bias_gelu(y, bias)
    return eager.linear_act(x, weight, bias, act)


def embedding_fused(
    input_ids, token_type_ids, position_ids,
    word_emb, pos_emb, type_emb, ln_weight, ln_bias, eps: float = 1e-12,
):
    if _use_hip(word_emb):
        from .functions import EmbeddingFusedFn

        return EmbeddingFusedFn.apply(
            input_ids, token_type_ids, position_ids,
            word_emb, pos_emb, type_emb, ln_weight, ln_bias, eps,
        )
    return eager.embedding_fused(
        input_ids, token_type_ids, position_ids,
        word_emb, pos_emb, type_emb, ln_weight, ln_bias, eps,
    )


def sgd_step(params, grads, lr, momentum=0.0, weight_decay=0.0,
             momentum_bufs=None, master_params=None):
    """Multi-tensor SGD step; fused HIP path lives in optim.FusedSGD."""
    eager.sgd_step(params, grads, lr, momentum, weight_decay, momentum_bufs, master_params)
