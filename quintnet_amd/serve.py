"""Minimal single-GPU inference server over the KV-cached generate path.

    python -m quintnet_amd.serve --checkpoint merged.pt --port 8000
    curl -X POST localhost:8000/generate -H 'content-type: application/json' \
         -d '{"input_ids": [1, 2, 3], "max_new_tokens": 16}'

Token-id interface (no tokenizer dependency baked in); temperature/top-k
sampling and int8 KV cache are exposed per request.  Beyond reference
parity (the reference has no serving path).
"""

from __future__ import annotations

import argparse
from typing import Optional

import torch


def build_app(stage, draft=None):
    """FastAPI app over ``stage``; pass ``draft`` (a smaller GPT2Stage
    sharing the vocab) to serve speculative decoding on requests that
    set ``"speculative": true``."""
    from fastapi import Body, FastAPI

    app = FastAPI(title="quintnet_amd GPT-2 server")

    @app.get("/health")
    def health():
        return {"status": "ok", "device": str(next(stage.parameters()).device)}

    @app.post("/generate")
    def generate(req: dict = Body(...)):
        dev = next(stage.parameters()).device
        ids = torch.tensor([req["input_ids"]], dtype=torch.long, device=dev)
        if int(req.get("num_beams", 0)) > 1:
            from .models import beam_search

            out = beam_search(
                stage, ids,
                max_new_tokens=int(req.get("max_new_tokens", 32)),
                num_beams=int(req.get("num_beams")),
                length_penalty=float(req.get("length_penalty", 1.0)),
                eos_token_id=req.get("eos_token_id"),
            )
            return {
                "output_ids": out[0].tolist(),
                "new_ids": out[0, ids.shape[1]:].tolist(),
            }
        if draft is not None and req.get("speculative"):
            from .models import speculative_generate

            out = speculative_generate(
                stage, draft, ids,
                max_new_tokens=int(req.get("max_new_tokens", 32)),
                draft_k=int(req.get("draft_k", 4)),
                temperature=float(req.get("temperature", 0.0)),
                top_k=int(req.get("top_k", 0)),
                top_p=float(req.get("top_p", 0.0)),
                eos_token_id=req.get("eos_token_id"),
                seed=req.get("seed"),
            )
            return {
                "output_ids": out[0].tolist(),
                "new_ids": out[0, ids.shape[1]:].tolist(),
            }
        out = stage.generate(
            ids,
            max_new_tokens=int(req.get("max_new_tokens", 32)),
            temperature=float(req.get("temperature", 0.0)),
            top_k=int(req.get("top_k", 0)),
            top_p=float(req.get("top_p", 0.0)),
            repetition_penalty=float(req.get("repetition_penalty", 1.0)),
            eos_token_id=req.get("eos_token_id"),
            cache_dtype=req.get("cache_dtype"),
        )
        return {
            "output_ids": out[0].tolist(),
            "new_ids": out[0, ids.shape[1]:].tolist(),
        }

    return app


def _load_stage(checkpoint: Optional[str], tiny: bool):
    from .models import GPT2Config, GPT2Stage

    dev = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    if tiny or checkpoint is None:
        cfg = GPT2Config(vocab_size=512, n_positions=256, n_embd=64,
                         n_layer=2, n_head=2, dropout=0.0)
    else:
        cfg = GPT2Config(dropout=0.0)
    stage = GPT2Stage(cfg, pp_rank=0, pp_size=1, tp_group=None,
                      device=dev, dtype=dtype)
    if checkpoint:
        sd = torch.load(checkpoint, map_location="cpu", weights_only=False)
        sd = sd.get("model_state_dict", sd)
        stage.load_state_dict(sd, strict=False)
    stage.eval()
    return stage


def main():
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--tiny", action="store_true")
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    args = ap.parse_args()
    app = build_app(_load_stage(args.checkpoint, args.tiny))
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
