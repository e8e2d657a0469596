"""Minimal generation server over the native decode engine.

Production-serving entry for trained checkpoints: the same hipGraph-captured
KV-cache decode path the RL rollout loop uses (models/nn/generation.py),
behind a small HTTP API.  The reference has no serving story of its own
(its hh example talks to an external Triton server); this closes the loop
for deploying `save_pretrained` checkpoints.

    python -m trlx_amd.serve --model ckpts/.../hf_model [--port 8720]

API:
    GET  /health                      -> {"ok": true, "model": "..."}
    POST /generate {"prompts": [...], "max_new_tokens": 40,
                    "temperature": 1.0, "top_k": 0, "top_p": 1.0,
                    "do_sample": true}
                                      -> {"completions": ["...", ...]}
"""

import argparse
from typing import List, Optional

import torch
from fastapi import FastAPI
from pydantic import BaseModel

from .models.modeling_base import PreTrainedModelWrapper
from .models.nn.generation import GenerateConfig, generate
from .utils.tokenizer import get_tokenizer


class GenerateRequest(BaseModel):
    prompts: List[str]
    max_new_tokens: int = 40
    temperature: float = 1.0
    top_k: int = 0
    top_p: float = 1.0
    do_sample: bool = True
    seed: Optional[int] = None


def create_app(model, tokenizer, device=None, continuous_slots: int = 0,
               cache_len: int = 2048) -> FastAPI:
    """Wrap a loaded model (wrapper or bare CausalTransformer) + tokenizer.

    ``continuous_slots > 0`` serves through the continuous batcher
    (trlx_amd/serving.py): concurrent requests share one slot pool and
    admit/retire per token step instead of padding into one fixed batch."""
    app = FastAPI()
    base = model.base_model if hasattr(model, "base_model") else model
    if device is None:
        device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    base = base.to(device).eval()
    if device.type == "cuda":
        base.to(torch.bfloat16)
    name = getattr(getattr(base, "config", None), "arch_name", "model")

    @app.get("/health")
    def health():
        return {"ok": True, "model": name, "device": str(device)}

    batcher = None
    if continuous_slots > 0:
        from .serving import ContinuousBatcher

        batcher = ContinuousBatcher(base, slots=continuous_slots, cache_len=cache_len,
                                    gen=GenerateConfig(
                                        do_sample=True,
                                        eos_token_id=tokenizer.eos_token_id)).start()
        app.state.batcher = batcher

    @app.post("/generate")
    def gen(req: GenerateRequest):
        if batcher is not None:
            # unpadded per-prompt admission; futures resolve as slots finish
            futs = [batcher.submit(tokenizer(p, return_tensors="pt")["input_ids"][0],
                                   max_new_tokens=req.max_new_tokens,
                                   temperature=req.temperature,
                                   do_sample=req.do_sample)
                    for p in req.prompts]
            return {"completions": [
                tokenizer.decode(f.result(), skip_special_tokens=True) for f in futs]}
        tokenizer.padding_side = "left"
        enc = tokenizer(req.prompts, padding=True, return_tensors="pt")
        ids = enc["input_ids"].to(device)
        mask = enc["attention_mask"].to(device)
        cfg = GenerateConfig(
            max_new_tokens=req.max_new_tokens, do_sample=req.do_sample,
            temperature=req.temperature, top_k=req.top_k, top_p=req.top_p,
            eos_token_id=tokenizer.eos_token_id, pad_token_id=tokenizer.pad_token_id,
            seed=req.seed,
        )
        with torch.no_grad():
            out = generate(base, ids, mask, gen=cfg)
        completions = [
            tokenizer.decode(row[ids.shape[1]:], skip_special_tokens=True)
            for row in out.cpu()
        ]
        return {"completions": completions}

    return app


def load(model_path: str, tokenizer_path: Optional[str] = None):
    model = PreTrainedModelWrapper.from_pretrained(model_path)
    tok = get_tokenizer(tokenizer_path or model_path)
    return model, tok


def main():
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--model", required=True, help="HF-format dir (save_pretrained output) or preset")
    p.add_argument("--tokenizer", default=None)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8720)
    p.add_argument("--slots", type=int, default=0,
                   help="enable continuous batching with this many KV slots")
    p.add_argument("--cache-len", type=int, default=2048)
    args = p.parse_args()
    model, tok = load(args.model, args.tokenizer)
    uvicorn.run(create_app(model, tok, continuous_slots=args.slots,
                           cache_len=args.cache_len),
                host=args.host, port=args.port)


if __name__ == "__main__":
    main()
