"""Offline ZeRO-1 checkpoint merge/convert CLI.

Role parity with the reference's ``optimizer/convert_zero_checkpoints.py``
(:15-179, console script ``nxd_convert_zero_checkpoints``, setup.py:66):
take a training checkpoint directory whose ``optim/`` holds one ZeRO-1
shard file per data-parallel rank, and merge the dp shards OFFLINE (no
process group needed) into one consolidated optimizer state per
model-parallel (tp, pp[, ep/cp]) slice:

    python -m neuronx_distributed_amd.scripts.convert_zero_checkpoints \
        --input  /ckpts/run1/step_1000  --output /ckpts/run1/merged

Output: ``{output}/optim_full_{mp_suffix}.pt`` per slice with
``{"state": {param_idx: {"master": fp32 tensor, "exp_avg": ..,
"exp_avg_sq": .., "step": n}}, "shapes": [...]}`` — per-parameter fp32
master weights and Adam moments, ready for conversion to a fresh
optimizer or for weight extraction.
"""

import argparse
import os
import re
from collections import defaultdict
from typing import Dict, List

import torch


_SHARD_RE = re.compile(
    r"dp_rank_(\d+)"
    r"(?:_cp_rank_(\d+))?"
    r"(?:_ep_rank_(\d+))?"
    r"_tp_rank_(\d+)_pp_rank_(\d+)\.pt$")


def _load_maybe_xser(optim_dir: str, fname: str):
    """Load a shard file, resolving the xser per-tensor format when its
    ``.info.pt`` index is present (trainer/checkpoint.py xser mode)."""
    full = os.path.join(optim_dir, fname)
    if os.path.exists(full + ".info.pt"):
        from ..trainer.checkpoint import _xser_unflatten

        skeleton = torch.load(full, map_location="cpu", weights_only=False)
        return _xser_unflatten(
            skeleton,
            lambda tid: torch.load(
                os.path.join(optim_dir, f"{fname}.tensors/tensor_{tid}.pt"),
                map_location="cpu", weights_only=False))
    return torch.load(full, map_location="cpu", weights_only=False)


def merge_zero_shards(shards: List[dict]) -> dict:
    """Merge one model-parallel slice's dp-rank shards (ascending dp rank)
    into consolidated per-parameter state."""
    shards = sorted(shards, key=lambda s: [m["rank"]
                                           for m in s["shard_meta"]])
    first = shards[0]
    n_buckets = len(first["shard_meta"])
    out_state: Dict[int, dict] = {}
    pidx = 0
    step = first.get("step_count", 0)
    base_sds = [s["base_optimizer"] for s in shards]
    for bi in range(n_buckets):
        metas = [s["shard_meta"][bi] for s in shards]
        order = sorted(range(len(shards)), key=lambda i: metas[i]["rank"])
        world = metas[0]["world"]
        assert len({m["rank"] for m in metas}) == len(metas), \
            f"duplicate dp shard ranks in bucket {bi}"
        assert len(metas) == world, (
            f"bucket {bi}: found {len(metas)} shards, expected {world}")

        def cat(field_get):
            return torch.cat([field_get(order[r]) for r in range(world)])

        masters = [s["masters"][bi] if s.get("masters") else None
                   for s in shards]
        if any(m is None for m in masters):
            raise ValueError(
                "shards were saved without master weights "
                "(use_master_weights_in_ckpt=False) — nothing to merge")
        full_master = cat(lambda i: masters[i])

        # Adam moments: fused kernel state or base-optimizer state
        fused = [s["fused_state"][bi] if s.get("fused_state") else None
                 for s in shards]
        if all(f is not None for f in fused):
            full_m = cat(lambda i: fused[i]["m"])
            full_v = cat(lambda i: fused[i]["v"])
        else:
            # base optimizer param order == bucket order within its group
            def moment(i, key):
                st = base_sds[i]["state"].get(bi, {})
                return st.get(key)

            if all(moment(i, "exp_avg") is not None
                   for i in range(len(shards))):
                full_m = cat(lambda i: moment(i, "exp_avg"))
                full_v = cat(lambda i: moment(i, "exp_avg_sq"))
                steps = moment(0, "step")
                if steps is not None:
                    step = int(steps) if not isinstance(steps, torch.Tensor) \
                        else int(steps.item())
            else:
                full_m = full_v = None

        for (s0, e0, shape) in metas[0]["segments"]:
            ps = {"master": full_master[s0:e0].reshape(shape), "step": step}
            if full_m is not None:
                ps["exp_avg"] = full_m[s0:e0].reshape(shape)
                ps["exp_avg_sq"] = full_v[s0:e0].reshape(shape)
            out_state[pidx] = ps
            pidx += 1
    return {"state": out_state,
            "shapes": [p["master"].shape for p in out_state.values()]}


def convert(input_dir: str, output_dir: str) -> List[str]:
    optim_dir = os.path.join(input_dir, "optim")
    if not os.path.isdir(optim_dir):
        raise FileNotFoundError(f"{optim_dir} does not exist")
    groups = defaultdict(list)  # mp-suffix -> [fname]
    for f in sorted(os.listdir(optim_dir)):
        m = _SHARD_RE.match(f)
        if m:
            dp, cp, ep, tp, pp = m.groups()
            # cp ranks are part of the zero1 sharding dim (merged dp x cp),
            # so they belong to the SAME merge group, not the key
            key = f"tp_{tp}_pp_{pp}" + (f"_ep_{ep}" if ep else "")
            groups[key].append(f)
    if not groups:
        raise FileNotFoundError(f"no zero1 shard files under {optim_dir}")
    os.makedirs(output_dir, exist_ok=True)
    written = []
    for key, files in sorted(groups.items()):
        shards = [_load_maybe_xser(optim_dir, f) for f in files]
        # unwrap NxDOptimizer-style nesting if present
        shards = [s.get("optimizer", s) if isinstance(s, dict) else s
                  for s in shards]
        merged = merge_zero_shards(shards)
        out = os.path.join(output_dir, f"optim_full_{key}.pt")
        torch.save(merged, out)
        written.append(out)
        print(f"wrote {out}: {len(merged['state'])} params")
    return written


def main():
    ap = argparse.ArgumentParser(
        description="Merge per-dp-rank ZeRO-1 optimizer shards offline")
    ap.add_argument("--input", required=True,
                    help="checkpoint tag dir containing optim/")
    ap.add_argument("--output", required=True, help="output directory")
    args = ap.parse_args()
    convert(args.input, args.output)


if __name__ == "__main__":
    main()
