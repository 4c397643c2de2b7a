#!/usr/bin/env python3
"""Minimal serving front-end: FastAPI over the generation engine
(hipGraph decode on GPU).  Single-process TP=1 demo; for TP>1 run the
engine ranks with torchrun and put the server on rank 0's feed.

  python examples/inference/serve.py --model llama3-8b --port 8000
  curl -X POST localhost:8000/generate -H 'Content-Type: application/json' \
       -d '{"token_ids": [[1, 2, 3]], "max_new_tokens": 16}'
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))


def build_app(model_name: str = "test-d128"):
    import torch
    import torch.distributed as dist
    from fastapi import FastAPI
    from pydantic import BaseModel

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29794")
    os.environ.setdefault("NXDA_FAST_INIT", "1")
    on_gpu = torch.cuda.is_available()
    if not dist.is_initialized():
        dist.init_process_group("nccl" if on_gpu else "gloo", rank=0,
                                world_size=1)

    from neuronx_distributed_amd.parallel import parallel_state as ps
    from neuronx_distributed_amd.models import get_config, LlamaForCausalLM
    from neuronx_distributed_amd.inference.generation import generate
    from neuronx_distributed_amd.utils.sampling import Sampler

    if not ps.model_parallel_is_initialized():
        ps.initialize_model_parallel(tensor_model_parallel_size=1)
    cfg = get_config(model_name)
    device = torch.device("cuda") if on_gpu else torch.device("cpu")
    prev = torch.get_default_dtype()
    if on_gpu:
        torch.set_default_dtype(torch.bfloat16)
    torch.manual_seed(0)
    with torch.device(device):
        model = LlamaForCausalLM(cfg).eval()
    torch.set_default_dtype(prev)

    app = FastAPI(title="neuronx-distributed-amd serving demo")

    class GenRequest(BaseModel):
        token_ids: list  # List[List[int]] — bring your own tokenizer
        max_new_tokens: int = 32
        do_sample: bool = False
        top_k: int = 50
        temperature: float = 1.0

    @app.post("/generate")
    def gen(req: GenRequest):
        x = torch.tensor(req.token_ids, dtype=torch.long, device=device)
        sampler = Sampler(do_sample=req.do_sample, top_k=req.top_k,
                          temperature=req.temperature)
        with torch.no_grad():
            out = generate(model, x, max_new_tokens=req.max_new_tokens,
                           sampler=sampler)
        return {"token_ids": out[:, x.shape[1]:].tolist()}

    @app.get("/health")
    def health():
        return {"status": "ok", "model": model_name,
                "device": str(device)}

    return app


def main():
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama3-8b")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()
    uvicorn.run(build_app(args.model), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
