"""Out-of-band reward server (parity: reference examples/hh/ppo_hh.py:9,109-120
+ to_triton.py — trlX scores HH rollouts against a reward model served by
Triton over gRPC).  MI355X-native analog: a FastAPI/uvicorn HTTP service
hosting the reward model on its own GPU/process so rollout generation and
reward scoring overlap across processes.

    python reward_server.py [--rm-dir ckpts/.../rm] [--port 8710]

POST /reward  {"samples": ["...", ...]} -> {"scores": [float, ...]}
GET  /health

Without a trained RM checkpoint it serves the synthetic HH oracle
(helpfulness word score), which is what the offline example trains against.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from fastapi import FastAPI
from pydantic import BaseModel

from hh_task import oracle_reward

app = FastAPI()
_rm = {"model": None, "tok": None}


class RewardRequest(BaseModel):
    samples: list


@app.get("/health")
def health():
    return {"ok": True, "rm": _rm["model"] is not None}


@app.post("/reward")
def reward(req: RewardRequest):
    if _rm["model"] is not None:
        import torch

        tok, model = _rm["tok"], _rm["model"]
        ids = [tok._encode_one(s, 256, truncation=True) + [tok.eos_token_id]
               for s in req.samples]
        width = max(len(i) for i in ids)
        ids = [i + [tok.pad_token_id] * (width - len(i)) for i in ids]
        device = next(model.parameters()).device
        with torch.no_grad():
            scores = model.score(torch.tensor(ids, dtype=torch.long, device=device))
        return {"scores": scores.cpu().tolist()}
    return {"scores": oracle_reward(req.samples)}


def load_rm(rm_dir: str):
    import torch

    sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                    "..", "summarize_rlhf"))
    from reward_model import RewardModel
    from trlx_amd.utils.tokenizer import ByteTokenizer

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    _rm["model"] = RewardModel.load_checkpoint(rm_dir, device=device).eval()
    _rm["tok"] = ByteTokenizer()


if __name__ == "__main__":
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--rm-dir", default=None)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8710)
    args = p.parse_args()
    if args.rm_dir and os.path.isdir(args.rm_dir):
        load_rm(args.rm_dir)
    uvicorn.run(app, host=args.host, port=args.port)
