from .tp import build_tp_model, shard_llama_weights, TPEngineGroup


def build_ep_model(*a, **kw):  # lazy: ep pulls torch.distributed
    from .ep import build_ep_model as f
    return f(*a, **kw)


__all__ = ["build_tp_model", "shard_llama_weights", "TPEngineGroup",
           "build_ep_model"]
