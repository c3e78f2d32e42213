"""Minimal inference service over the KV-cached Llama decoder.

Deployment-side complement to the training engine: load a rank-0
checkpoint produced by the training loop (the plain torch.save contract)
and serve token-level generation over HTTP. Single process, one GPU (or
CPU); data parallelism is a training concern — serve replicas behind any
load balancer instead.

    python examples/serve_llama.py --config tiny --port 8000
    curl -s localhost:8000/generate -d '{"tokens": [1,2,3], "max_new_tokens": 8}'
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from fastapi import FastAPI
from pydantic import BaseModel, Field


class GenerateRequest(BaseModel):
    tokens: list[int] = Field(min_length=1)
    max_new_tokens: int = Field(default=16, ge=1, le=4096)
    temperature: float = Field(default=0.0, ge=0.0, le=10.0)
    top_k: int | None = Field(default=None, ge=1)


def build_app(model=None, config="tiny", checkpoint=None):
    from axon.models.llama import (LlamaForCausalLM, llama3_8b_config,
                                   llama3_1b_config, llama_tiny_config)

    if model is None:
        cfg = {"tiny": llama_tiny_config, "llama3-1b": llama3_1b_config,
               "llama3-8b": llama3_8b_config}[config]()
        device = "cuda" if torch.cuda.is_available() else "cpu"
        dtype = torch.bfloat16 if device == "cuda" else torch.float32
        with torch.device(device):
            model = LlamaForCausalLM(cfg)
        model = model.to(dtype=dtype).eval()
        if checkpoint:
            ck = torch.load(checkpoint, map_location=device,
                            weights_only=True)
            model.load_state_dict(ck["model"] if "model" in ck else ck)

    app = FastAPI(title="axon llama service")
    dev = next(model.parameters()).device

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(dev),
                "params": sum(p.numel() for p in model.parameters())}

    @app.post("/generate")
    def generate(req: GenerateRequest):
        ids = torch.tensor([req.tokens], dtype=torch.long, device=dev)
        out = model.generate(ids, req.max_new_tokens,
                             temperature=req.temperature, top_k=req.top_k)
        return {"tokens": out[0].tolist(),
                "new_tokens": out[0, ids.shape[1]:].tolist()}

    return app


def main():
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="tiny",
                    choices=["tiny", "llama3-1b", "llama3-8b"])
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args()
    uvicorn.run(build_app(config=args.config, checkpoint=args.checkpoint),
                host="127.0.0.1", port=args.port)


if __name__ == "__main__":
    main()
