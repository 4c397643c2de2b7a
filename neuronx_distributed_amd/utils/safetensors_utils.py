"""Shared-tensor-safe safetensors save (reference utils/safetensors_utils.py):
safetensors refuses aliased storages (tied embeddings); dedup before save and
re-tie on load."""

from typing import Dict, List

import torch


def find_shared_tensors(state_dict: Dict[str, torch.Tensor]) -> List[List[str]]:
    by_storage = {}
    for k, v in state_dict.items():
        if isinstance(v, torch.Tensor):
            ptr = v.untyped_storage().data_ptr()
            by_storage.setdefault(ptr, []).append(k)
    return [ks for ks in by_storage.values() if len(ks) > 1]


def dedup_state_dict(state_dict: Dict[str, torch.Tensor]):
    """Keep one name per shared storage; return (deduped, alias_map)."""
    groups = find_shared_tensors(state_dict)
    alias = {}
    out = dict(state_dict)
    for ks in groups:
        keep = sorted(ks)[0]
        for k in ks:
            if k != keep:
                del out[k]
                alias[k] = keep
    return out, alias


def save_safetensors(state_dict, path: str, metadata=None):
    from safetensors.torch import save_file

    deduped, alias = dedup_state_dict(state_dict)
    meta = dict(metadata or {})
    if alias:
        import json

        meta["nxd_amd_aliases"] = json.dumps(alias)
    save_file({k: v.contiguous() for k, v in deduped.items()}, path,
              metadata=meta)


def load_safetensors(path: str):
    import json

    from safetensors import safe_open
    from safetensors.torch import load_file

    sd = load_file(path)
    with safe_open(path, framework="pt") as f:
        meta = f.metadata() or {}
    for k, src in json.loads(meta.get("nxd_amd_aliases", "{}")).items():
        sd[k] = sd[src]
    return sd
