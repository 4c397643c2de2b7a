"""Load a HuggingFace llama-family checkpoint into the native model and
generate — the migration path for users coming from HF/the reference:

  torchrun --nproc-per-node 1 examples/inference/load_hf_checkpoint.py

Runs standalone on CPU with a tiny random-init HF model (no network);
point --hf-dir at a real downloaded HF checkpoint on a GPU box."""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

import torch
import torch.distributed as dist


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--hf-dir", default=None,
                   help="HF checkpoint dir (default: tiny random model)")
    p.add_argument("--prompt-len", type=int, default=16)
    p.add_argument("--new", type=int, default=8)
    args = p.parse_args()

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29731")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend)

    from neuronx_distributed_amd.inference import generate
    from neuronx_distributed_amd.models import LlamaForCausalLM, get_config
    from neuronx_distributed_amd.overrides import convert_hf_llama_state_dict
    from neuronx_distributed_amd.parallel import parallel_state as ps

    ps.initialize_model_parallel(tensor_model_parallel_size=1)

    if args.hf_dir:
        from transformers import AutoConfig, AutoModelForCausalLM

        hf = AutoModelForCausalLM.from_pretrained(args.hf_dir,
                                                  dtype=torch.bfloat16)
        hc = AutoConfig.from_pretrained(args.hf_dir)
        cfg = get_config(
            "llama2-7b", hidden_size=hc.hidden_size,
            intermediate_size=hc.intermediate_size,
            num_hidden_layers=hc.num_hidden_layers,
            num_attention_heads=hc.num_attention_heads,
            num_key_value_heads=hc.num_key_value_heads,
            vocab_size=hc.vocab_size, rope_theta=hc.rope_theta,
            max_position_embeddings=hc.max_position_embeddings,
            rope_scaling=getattr(hc, "rope_scaling", None))
    else:  # offline demo: tiny random-init HF model
        from transformers import LlamaConfig as HFConfig
        from transformers import LlamaForCausalLM as HFModel

        torch.manual_seed(0)
        hf = HFModel(HFConfig(hidden_size=64, intermediate_size=128,
                              num_hidden_layers=2, num_attention_heads=4,
                              num_key_value_heads=2, vocab_size=256,
                              max_position_embeddings=128,
                              tie_word_embeddings=False))
        cfg = get_config("tiny")

    model = LlamaForCausalLM(cfg)
    missing, unexpected = model.load_state_dict(
        convert_hf_llama_state_dict(hf.state_dict()), strict=False)
    assert not unexpected, unexpected
    del hf
    if torch.cuda.is_available():
        model = model.cuda().bfloat16()
    model.eval()

    torch.manual_seed(1)
    x = torch.randint(0, cfg.vocab_size, (1, args.prompt_len))
    if torch.cuda.is_available():
        x = x.cuda()
    out = generate(model, x, max_new_tokens=args.new)
    print("generated token ids:", out[0, -args.new:].tolist())
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
