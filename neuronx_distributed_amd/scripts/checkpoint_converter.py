"""HF <-> sharded checkpoint converter (reference
scripts/checkpoint_converter.py:23-90 ``CheckpointConverterBase``):
full->sharded and sharded->full across TP/PP, driven by per-layer
partition-dim maps; plus the zero1 dp-shard merge CLI (reference
optimizer/convert_zero_checkpoints.py:15-179)."""

import argparse
import os
from typing import Dict, List, Optional

import torch

from ..parallel.utils import create_local_weight


class CheckpointConverterBase:
    """Subclass and override the *_partition_dim tables per architecture.
    Keys are SUFFIX matches on parameter names.

    Limitation: GQA KV-head REPLICATION (tp > kv heads) is a runtime
    resharding concern — load the FULL checkpoint through
    ``parallel.checkpointing.load(sharded=False)`` whose preshard_hook
    applies the replication layout; this offline converter shards evenly
    and covers tp <= kv_heads."""

    # name-suffix -> partition dim (column-parallel: 0, row-parallel: 1);
    # 3-D expert-fused weights (E, in, out) map column->dim 2, row->dim 1
    COLUMN_PARALLEL_SUFFIXES = ["q_proj.weight", "k_proj.weight",
                                "v_proj.weight", "gate_proj.weight",
                                "up_proj.weight", "lm_head.weight",
                                "embed_tokens.weight", "wte.weight",
                                "c_fc.weight",
                                # native layer names (this package)
                                "qkv_proj.weight_q", "qkv_proj.weight_k",
                                "qkv_proj.weight_v", "qkv_proj.bias_q",
                                "qkv_proj.bias_k", "qkv_proj.bias_v",
                                "embed_in.weight",
                                "embed_out.weight", "dense_h_to_4h.weight"]
    ROW_PARALLEL_SUFFIXES = ["o_proj.weight", "down_proj.weight",
                             "c_proj.weight", "dense_4h_to_h.weight",
                             "attention.dense.weight"]
    # fused [gate; up] / [q;k;v] weights: (suffix, num_blocks)
    STRIDED_COLUMN_SUFFIXES = [("gate_up_proj.weight", 2),
                               ("c_attn.weight", 3),
                               ("query_key_value.weight", 3)]

    def _dim_of(self, name: str):
        for suf, stride in self.STRIDED_COLUMN_SUFFIXES:
            if name.endswith(suf):
                return 0, stride
        for suf in self.COLUMN_PARALLEL_SUFFIXES:
            if name.endswith(suf):
                return 0, 1
        for suf in self.ROW_PARALLEL_SUFFIXES:
            if name.endswith(suf):
                return 1, 1
        return None, 1

    # -- full -> sharded ---------------------------------------------------
    def shard_full_checkpoint(self, full_sd: Dict[str, torch.Tensor],
                              tp_degree: int) -> List[Dict[str, torch.Tensor]]:
        shards = []
        for r in range(tp_degree):
            shard = {}
            for name, w in full_sd.items():
                dim, stride = self._dim_of(name)
                if dim is None or not isinstance(w, torch.Tensor):
                    shard[name] = w
                else:
                    if w.dim() == 3:  # expert-fused (E, in, out)
                        dim = 2 if dim == 0 else 1
                    per = w.shape[dim] // tp_degree
                    shard[name] = create_local_weight(w, dim, per, stride,
                                                      rank=r,
                                                      world_size=tp_degree)
            shards.append(shard)
        return shards

    # -- sharded -> full ---------------------------------------------------
    def merge_sharded_checkpoints(self, shards: List[Dict[str, torch.Tensor]]
                                  ) -> Dict[str, torch.Tensor]:
        tp = len(shards)
        full = {}
        for name, w0 in shards[0].items():
            dim, stride = self._dim_of(name)
            if dim is None or not isinstance(w0, torch.Tensor):
                full[name] = w0
                continue
            parts = [s[name] for s in shards]
            if w0.dim() == 3:  # expert-fused (E, in, out)
                dim = 2 if dim == 0 else 1
            if stride == 1:
                full[name] = torch.cat(parts, dim=dim)
            else:
                # each rank holds [b0_r | b1_r | ...]: regroup per block
                blocks = [p.chunk(stride, dim=dim) for p in parts]
                full[name] = torch.cat(
                    [torch.cat([blocks[r][b] for r in range(tp)], dim=dim)
                     for b in range(stride)], dim=dim)
        return full

    # -- file-level driver -------------------------------------------------
    def convert_from_full_state(self, input_path: str, output_dir: str,
                                tp_degree: int, pp_degree: int = 1):
        full = torch.load(input_path, map_location="cpu", weights_only=False)
        shards = self.shard_full_checkpoint(full, tp_degree)
        os.makedirs(output_dir, exist_ok=True)
        for r, shard in enumerate(shards):
            torch.save(shard, os.path.join(
                output_dir, f"tp_rank_{r:02d}_pp_rank_00.pt"))

    def convert_to_full_state(self, input_dir: str, output_path: str,
                              tp_degree: int):
        shards = [
            torch.load(os.path.join(input_dir,
                                    f"tp_rank_{r:02d}_pp_rank_00.pt"),
                       map_location="cpu", weights_only=False)
            for r in range(tp_degree)
        ]
        torch.save(self.merge_sharded_checkpoints(shards), output_path)


def merge_zero_checkpoints(ckpt_dir: str, tag: str,
                           output_path: Optional[str] = None):
    """Offline merge of per-DP-rank ZeRO-1 optimizer shards (reference
    optimizer/convert_zero_checkpoints.py): reassembles each bucket's fp32
    master from the rank shards."""
    optim_dir = os.path.join(ckpt_dir, str(tag), "optim")
    files = sorted(os.listdir(optim_dir))
    states = [torch.load(os.path.join(optim_dir, f), map_location="cpu",
                         weights_only=False) for f in files]
    merged = {"masters": []}
    n_buckets = len(states[0]["masters"])
    for b in range(n_buckets):
        shards = [s["masters"][b] for s in states]
        merged["masters"].append(torch.cat(shards))
    if output_path:
        torch.save(merged, output_path)
    return merged


def main():
    p = argparse.ArgumentParser(description="nxd-amd checkpoint converter")
    p.add_argument("--mode", choices=["full2sharded", "sharded2full",
                                      "merge-zero"], required=True)
    p.add_argument("--input", required=True)
    p.add_argument("--output", required=True)
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--tag", default="0")
    args = p.parse_args()
    c = CheckpointConverterBase()
    if args.mode == "full2sharded":
        c.convert_from_full_state(args.input, args.output, args.tp)
    elif args.mode == "sharded2full":
        c.convert_to_full_state(args.input, args.output, args.tp)
    else:
        merge_zero_checkpoints(args.input, args.tag, args.output)


if __name__ == "__main__":
    main()


def convert_hf_llama_state_dict(hf_sd):
    """HF LlamaForCausalLM checkpoint -> this package's native llama
    layout: q/k/v_proj become qkv_proj.weight_q/k/v, gate_proj+up_proj
    fuse into gate_up_proj.weight ([gate; up] rows — the stride-2 fused
    FULL layout the converter shards), rotary inv_freq buffers drop
    (recomputed at init).  Norms/o_proj/down_proj/embed/lm_head names are
    already identical."""
    out = {}
    gates = {}
    ups = {}
    for k, v in hf_sd.items():
        if k.endswith("rotary_emb.inv_freq"):
            continue
        if k.endswith("self_attn.q_proj.weight"):
            out[k.replace("q_proj.weight", "qkv_proj.weight_q")] = v
        elif k.endswith("self_attn.k_proj.weight"):
            out[k.replace("k_proj.weight", "qkv_proj.weight_k")] = v
        elif k.endswith("self_attn.v_proj.weight"):
            out[k.replace("v_proj.weight", "qkv_proj.weight_v")] = v
        elif k.endswith("self_attn.q_proj.bias"):  # qwen2-style
            out[k.replace("q_proj.bias", "qkv_proj.bias_q")] = v
        elif k.endswith("self_attn.k_proj.bias"):
            out[k.replace("k_proj.bias", "qkv_proj.bias_k")] = v
        elif k.endswith("self_attn.v_proj.bias"):
            out[k.replace("v_proj.bias", "qkv_proj.bias_v")] = v
        elif k.endswith("mlp.gate_proj.weight"):
            gates[k.rsplit("gate_proj.weight", 1)[0]] = v
        elif k.endswith("mlp.up_proj.weight"):
            ups[k.rsplit("up_proj.weight", 1)[0]] = v
        else:
            out[k] = v
    assert set(gates) == set(ups), "gate/up projections must pair up"
    for base, g in gates.items():
        out[base + "gate_up_proj.weight"] = torch.cat([g, ups[base]], dim=0)
    return out


def convert_hf_mixtral_state_dict(hf_sd, num_experts: int):
    """HF MixtralForCausalLM -> native layout: per-expert w1/w3 (gate/up,
    (I,H)) stack-transpose into the expert-fused gate_up (E, H, 2I)
    [gate|up] on the last dim; w2 (down, (H,I)) into down (E, I, H);
    block_sparse_moe.gate -> router.linear_router (fp32).  Attention and
    norms follow the llama mapping."""
    out = convert_hf_llama_state_dict(
        {k: v for k, v in hf_sd.items() if ".block_sparse_moe." not in k})
    layers = {}
    for k, v in hf_sd.items():
        if ".block_sparse_moe." not in k:
            continue
        base, rest = k.split(".block_sparse_moe.", 1)
        layers.setdefault(base, {})[rest] = v
    for base, params in layers.items():
        if "gate.weight" in params:
            out[f"{base}.block_sparse_moe.router.linear_router.weight"] = \
                params["gate.weight"].float()
        gate_up = []
        down = []
        for e in range(num_experts):
            w1 = params[f"experts.{e}.w1.weight"]  # (I, H) gate
            w3 = params[f"experts.{e}.w3.weight"]  # (I, H) up
            w2 = params[f"experts.{e}.w2.weight"]  # (H, I) down
            gate_up.append(torch.cat([w1.t(), w3.t()], dim=1))  # (H, 2I)
            down.append(w2.t())  # (I, H)
        out[f"{base}.block_sparse_moe.expert_mlps.gate_up_proj.weight"] = \
            torch.stack(gate_up)
        out[f"{base}.block_sparse_moe.expert_mlps.down_proj.weight"] = \
            torch.stack(down)
    return out
