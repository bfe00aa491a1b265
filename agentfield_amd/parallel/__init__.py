from .tp import build_tp_model, shard_llama_weights, TPEngineGroup

__all__ = ["build_tp_model", "shard_llama_weights", "TPEngineGroup"]
