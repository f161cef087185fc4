"""Bridge between the models' legacy tuple KV caches and the installed
transformers' Cache objects (`generate` passes a DynamicCache).

The training path never uses caches; decode keeps the simple per-layer
(k, v) tuple layout internally and converts at the model boundary."""

from typing import Optional


def cache_to_legacy(past):
    """Cache object | legacy tuple | None -> legacy tuple | None."""
    if past is None or isinstance(past, (tuple, list)):
        return past if past else None
    # a freshly-created DynamicCache may hold initialized-but-empty layers
    if len(past) == 0 or past.get_seq_length() == 0:
        return None
    if hasattr(past, "layers"):
        return tuple((layer.keys, layer.values) for layer in past.layers)
    # older transformers: DynamicCache exposes key_cache/value_cache lists
    return tuple(zip(past.key_cache, past.value_cache))


def cache_like(next_cache, template) -> Optional[object]:
    """Package per-layer (k, v) pairs in the caller's cache type: a Cache
    object when one was passed in, else a plain tuple."""
    if not next_cache:
        return None
    if template is None or isinstance(template, (tuple, list)):
        return tuple(next_cache)
    try:
        return type(template)(tuple(next_cache))
    except Exception:
        from transformers.cache_utils import DynamicCache
        return DynamicCache(tuple(next_cache))
