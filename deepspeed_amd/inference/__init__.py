"""Inference: TP-sharded serving with KV-cache decode over xGMI."""

from .engine import InferenceConfig, InferenceEngine
from .kv_cache import StaticKVCache

__all__ = ["InferenceEngine", "InferenceConfig", "StaticKVCache"]
