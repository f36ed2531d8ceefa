"""Serving front-end over the continuous batcher (reference: DeepSpeed-MII /
FastGen serving loop on top of inference/v2).

A FastAPI app exposing /generate: requests land in the ContinuousBatcher's
queue; a background scheduler thread runs step() continuously, batching
whatever is in flight. Each HTTP call waits on its request's completion
event — concurrent callers share decode batches automatically.
"""

import itertools
import threading
from typing import List, Optional

import torch

from .ragged import ContinuousBatcher, Request


class InferenceServer:
    def __init__(self, model, max_slots: int = 8, prefill_chunk: int = 0,
                 dtype=torch.float32, cache_cls=None):
        self.batcher = ContinuousBatcher(model, max_slots=max_slots,
                                         prefill_chunk=prefill_chunk,
                                         dtype=dtype, cache_cls=cache_cls)
        self._uid = itertools.count()
        self._events = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ------------------------------------------------------------- scheduler
    def _loop(self):
        while not self._stop.is_set():
            with self._lock:
                busy = self.batcher.has_work()
                finished = self.batcher.step() if busy else []
            for req in finished:
                self._events.pop(req.uid).set()
            if not busy:
                self._stop.wait(0.001)

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)

    # ------------------------------------------------------------------- api
    def submit(self, token_ids: List[int], max_new_tokens: int = 32,
               eos_token_id: Optional[int] = None, do_sample: bool = False,
               temperature: float = 1.0, top_k: int = 0,
               top_p: float = 1.0) -> Request:
        req = Request(uid=next(self._uid),
                      prompt=torch.as_tensor(token_ids, dtype=torch.long),
                      max_new_tokens=max_new_tokens,
                      eos_token_id=eos_token_id, do_sample=do_sample,
                      temperature=temperature, top_k=top_k, top_p=top_p)
        ev = threading.Event()
        with self._lock:
            self._events[req.uid] = ev
            self.batcher.put(req)
        req._event = ev
        return req

    def generate(self, token_ids: List[int], max_new_tokens: int = 32,
                 eos_token_id: Optional[int] = None,
                 timeout: float = 300.0, **sampling) -> List[int]:
        req = self.submit(token_ids, max_new_tokens, eos_token_id,
                          **sampling)
        if not req._event.wait(timeout):
            raise TimeoutError(f"request {req.uid} timed out")
        return req.generated


def build_app(server: InferenceServer):
    """FastAPI app: POST /generate {"token_ids": [...], "max_new_tokens": N}
    -> {"generated": [...]}; GET /health."""
    from fastapi import FastAPI
    from pydantic import BaseModel

    class GenRequest(BaseModel):
        token_ids: List[int]
        max_new_tokens: int = 32
        eos_token_id: Optional[int] = None
        do_sample: bool = False
        temperature: float = 1.0
        top_k: int = 0
        top_p: float = 1.0

    app = FastAPI(title="deepspeed_amd inference")

    @app.get("/health")
    def health():
        return {"status": "ok",
                "active": len(server.batcher.active),
                "pending": len(server.batcher.pending)}

    @app.post("/generate")
    def generate(r: GenRequest):
        out = server.generate(r.token_ids, r.max_new_tokens, r.eos_token_id,
                              do_sample=r.do_sample,
                              temperature=r.temperature,
                              top_k=r.top_k, top_p=r.top_p)
        return {"generated": out}

    return app


def main():
    """python -m deepspeed_amd.inference.server --model llama-mini"""
    import argparse

    import uvicorn

    from ..models import (LlamaForCausalLM, llama3_8b, llama_mini,
                          llama_tiny, phi3_mini)
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-mini",
                    choices=["llama3-8b", "llama-mini", "phi3-mini", "tiny"])
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-slots", type=int, default=16)
    args = ap.parse_args()
    cfg = {"llama3-8b": llama3_8b, "llama-mini": llama_mini,
           "phi3-mini": phi3_mini, "tiny": llama_tiny}[args.model]()
    model = LlamaForCausalLM(cfg)
    dtype = torch.bfloat16 if torch.cuda.is_available() else torch.float32
    model = model.to(dtype)
    if torch.cuda.is_available():
        model = model.cuda()
    server = InferenceServer(model, max_slots=args.max_slots,
                             dtype=dtype).start()
    uvicorn.run(build_app(server), host="127.0.0.1", port=args.port)


if __name__ == "__main__":
    main()
