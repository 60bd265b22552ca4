"""Minimal text-generation serving endpoint over a trained checkpoint.

Beyond the reference's scope (midGPT has only offline sample.py —
reference sample.py:29-37); included because this framework targets
production serving as well as pretraining. Loads a rundir checkpoint the
same way sample.py does, then serves batched KV-cache generation
(midgpt_amd/generate.py) over HTTP.

    python serve.py --ckpt_dir runs/xl [--host 127.0.0.1 --port 8000]

    POST /generate {"prompt": "...", "max_new_tokens": 200,
                    "temperature": 0.8, "num_samples": 1}
    GET  /healthz
"""
import argparse

import torch

from sample import load_model_and_tokenizer


def build_app(ckpt_dir: str, device: str | None = None):
    from fastapi import FastAPI
    from pydantic import BaseModel

    if device is None:
        device = "cuda" if torch.cuda.is_available() else "cpu"
    model, encode, decode, config = load_model_and_tokenizer(ckpt_dir, device)

    class GenRequest(BaseModel):
        prompt: str = "\n"
        max_new_tokens: int = 200
        temperature: float = 0.8
        num_samples: int = 1
        seed: int | None = None

    app = FastAPI(title="midgpt_amd")

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "device": device,
                "block_size": config.model_config.block_size}

    @app.post("/generate")
    def generate_ep(req: GenRequest):
        from midgpt_amd.generate import generate
        ids = encode(req.prompt)
        idx = torch.tensor([ids] * req.num_samples, dtype=torch.long,
                           device=device)
        gen = None
        if req.seed is not None:
            gen = torch.Generator().manual_seed(req.seed)
        out = generate(model, idx, req.max_new_tokens,
                       temperature=req.temperature, generator=gen)
        return {"samples": [decode(out[i].tolist())
                            for i in range(req.num_samples)]}

    return app


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ckpt_dir", required=True)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8000)
    args = p.parse_args()
    import uvicorn
    uvicorn.run(build_app(args.ckpt_dir), host=args.host, port=args.port)


if __name__ == "__main__":
    main()
