from .bert import BertConfig
from .bert_layers import (
    BertEmbeddings,
    BertLayerBody,
    BertLayerHead,
    BertLayerTail,
    BertPooler,
    BertSelfAttention,
    BertTailForClassification,
    LinearActivation,
    SkyLayerNorm,
    bert_pipeline_config,
)
from .resnet import ResHead, ResLayer, ResTail, resnet_pipeline_config

__all__ = [
    "BertConfig", "BertEmbeddings", "BertLayerHead", "BertLayerBody",
    "BertLayerTail", "BertPooler", "BertTailForClassification",
    "BertSelfAttention", "LinearActivation", "SkyLayerNorm",
    "bert_pipeline_config", "ResHead", "ResLayer", "ResTail",
    "resnet_pipeline_config",
]
