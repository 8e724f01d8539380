"""HTTP inference server: OpenAI-compatible completions over the
KV-cached GenerationEngine.

The reference ships only a terminal REPL (reference Chat.py:472-937); this
module adds the serving path a production deployment needs: an ASGI app
with /v1/completions, /v1/chat/completions (ChatML via the tokenizer's
conversation encoding), SSE streaming, /health, /v1/models and a
Prometheus /metrics endpoint. The default serving mode is token-level
continuous batching (inference/continuous.py — requests join the running
decode batch between steps; a lone request pays batch-1 compute);
serving="dynamic" keeps request-level batching with the hipGraph
single-stream path for streaming.

Run: python serve.py --checkpoint PATH [--port 8000] [--quantize int8]
"""

from __future__ import annotations

import json
import queue
import threading
import time
from typing import Dict, List, Optional

import torch

from .engine import GenerationConfig, GenerationEngine


def _gen_config(body: Dict) -> GenerationConfig:
    cfg = GenerationConfig.from_mode(body.get("mode", "standard"))
    if "max_tokens" in body:
        cfg.max_new_tokens = int(body["max_tokens"])
    if "temperature" in body:
        cfg.temperature = float(body["temperature"])
    if "top_p" in body:
        cfg.top_p = float(body["top_p"])
    if "top_k" in body:
        cfg.top_k = int(body["top_k"])
    if "repetition_penalty" in body:
        cfg.repetition_penalty = float(body["repetition_penalty"])
    if "stop_token_ids" in body:
        cfg.stop_token_ids = list(body["stop_token_ids"])
    return cfg


class _DynamicBatcher:
    """Dynamic request batching: non-streaming requests queue up and are
    decoded together through GenerationEngine.generate_batch (left-padded
    shared KV cache, per-row sampling configs). Requests that arrive while
    a batch is decoding form the next batch — throughput scales with
    concurrency without a latency penalty for the lone request."""

    def __init__(self, engine: GenerationEngine, max_batch: int = 8,
                 window_ms: float = 2.0, lock: Optional[threading.Lock] = None):
        self.engine = engine
        self.lock = lock or threading.Lock()
        self.max_batch = max_batch
        self.window_s = window_ms / 1000.0
        self.q: "queue.Queue" = queue.Queue()
        self.batches_run = 0
        self.rows_seen = 0
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def submit(self, prompt_ids: List[int],
               cfg: GenerationConfig) -> List[int]:
        done = threading.Event()
        slot: Dict = {"ids": prompt_ids, "cfg": cfg, "done": done}
        self.q.put(slot)
        done.wait()
        if "error" in slot:
            raise slot["error"]
        return slot["out"]

    def _loop(self):
        while True:
            batch = [self.q.get()]
            deadline = time.monotonic() + self.window_s
            while len(batch) < self.max_batch:
                remaining = deadline - time.monotonic()
                if remaining <= 0:
                    break
                try:
                    batch.append(self.q.get(timeout=remaining))
                except queue.Empty:
                    break
            try:
                with self.lock:
                    outs = self.engine.generate_batch(
                        [r["ids"] for r in batch], [r["cfg"] for r in batch])
                for r, out in zip(batch, outs):
                    r["out"] = out
            except Exception as e:  # noqa: BLE001
                for r in batch:
                    r["error"] = e
            finally:
                self.batches_run += 1
                self.rows_seen += len(batch)
                for r in batch:
                    r["done"].set()


class _ContinuousWorker:
    """Serving loop over ContinuousBatchingEngine: requests (streaming or
    not) join the running decode batch at token granularity; a lone
    request pays batch-1 compute (active-row compaction)."""

    def __init__(self, model, tokenizer, max_batch: int = 8,
                 max_len: int = 2048):
        from .continuous import ContinuousBatchingEngine
        self.eng = ContinuousBatchingEngine(model, tokenizer,
                                            max_batch=max_batch,
                                            max_len=max_len)
        self.q: "queue.Queue" = queue.Queue()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def submit(self, prompt_ids: List[int], cfg: GenerationConfig,
               cb=None) -> List[int]:
        evt = threading.Event()
        item = {"ids": prompt_ids, "cfg": cfg, "cb": cb, "evt": evt}
        self.q.put(item)
        evt.wait()
        if "err" in item:
            raise item["err"]
        return item["out"]

    def _loop(self):
        tracked: Dict[int, Dict] = {}
        pending: List[Dict] = []
        while True:
            # drain arrivals; block only when completely idle
            block = not tracked and not pending
            while True:
                try:
                    pending.append(self.q.get(block=block, timeout=None))
                    block = False
                except queue.Empty:
                    break
                if not block:
                    break
            while True:     # opportunistic non-blocking drain
                try:
                    pending.append(self.q.get_nowait())
                except queue.Empty:
                    break
            for item in list(pending):
                if not self.eng.can_admit(len(item["ids"])):
                    if self.eng.n_active() == 0:
                        item["err"] = RuntimeError(
                            "prompt exceeds the serving context window")
                        item["evt"].set()
                        pending.remove(item)
                    continue
                try:
                    b = self.eng.admit(item["ids"], item["cfg"],
                                       item.get("cb"))
                    tracked[b] = item
                except Exception as e:  # noqa: BLE001
                    item["err"] = e
                    item["evt"].set()
                pending.remove(item)
            if self.eng.n_active():
                try:
                    self.eng.step()
                except Exception as e:  # noqa: BLE001
                    for item in tracked.values():
                        item["err"] = e
                        item["evt"].set()
                    tracked.clear()
            for b, item in list(tracked.items()):
                s = self.eng.slots[b]
                if not s.active and s.result is not None:
                    item["out"] = s.result
                    item["evt"].set()
                    del tracked[b]


def create_app(model, tokenizer, model_name: str = "luminaai-amd",
               max_batch: int = 8, serving: str = "continuous",
               max_len: int = 2048):
    """Build the ASGI app around an already-loaded model. serving:
    "continuous" (token-level joins, default) or "dynamic" (request-level
    batching with the hipGraph single-stream path for streaming)."""
    from fastapi import FastAPI
    from fastapi.responses import JSONResponse, StreamingResponse

    app = FastAPI(title="LuminaAI-AMD", version="0.1")
    device = next(model.parameters()).device
    engine = GenerationEngine(model, tokenizer, device)
    lock = threading.Lock()
    if serving == "continuous":
        worker = _ContinuousWorker(model, tokenizer, max_batch=max_batch,
                                   max_len=max_len)
        batcher = None
    else:
        batcher = _DynamicBatcher(engine, max_batch=max_batch, lock=lock)
        worker = None
    started = time.time()
    counters = {"requests": 0, "tokens_out": 0}
    from ..monitoring.prometheus import make_server_metrics
    prom = make_server_metrics()

    def _account(endpoint: str, n_out: int, t0: float):
        counters["requests"] += 1
        counters["tokens_out"] += n_out
        if prom is not None:
            prom["requests"].labels(endpoint=endpoint).inc()
            prom["tokens"].inc(n_out)
            prom["latency"].observe(time.perf_counter() - t0)

    def _decode(prompt_ids: List[int], cfg: GenerationConfig,
                cb=None, endpoint: str = "completions") -> List[int]:
        t0 = time.perf_counter()
        if worker is not None:
            # continuous batching: streaming and non-streaming alike join
            # the running decode batch at token granularity
            out = worker.submit(prompt_ids, cfg, cb=cb)
        elif cb is None:
            # dynamic mode, non-streaming: requests share one batched pass
            out = batcher.submit(prompt_ids, cfg)
        else:
            # dynamic mode, streaming: single-sequence (hipGraph) decode
            with lock:
                out = engine.generate(prompt_ids, cfg, stream_callback=cb)
        _account(endpoint, len(out), t0)
        return out

    def _sse_stream(prompt_ids: List[int], cfg: GenerationConfig,
                    wrap, endpoint: str = "completions") -> StreamingResponse:
        q: "queue.Queue[Optional[int]]" = queue.Queue()

        def run():
            try:
                _decode(prompt_ids, cfg, cb=q.put, endpoint=endpoint)
            finally:
                q.put(None)

        threading.Thread(target=run, daemon=True).start()

        def gen():
            while True:
                tok = q.get()
                if tok is None:
                    break
                piece = tokenizer.decode([tok])
                yield f"data: {json.dumps(wrap(piece))}\n\n"
            yield "data: [DONE]\n\n"

        return StreamingResponse(gen(), media_type="text/event-stream")

    @app.get("/health")
    def health():
        d = {"status": "ok", "device": str(device), "serving": serving,
             "uptime_s": round(time.time() - started, 1), **counters}
        if worker is not None:
            d.update(worker.eng.stats)
        else:
            d.update(batches_run=batcher.batches_run,
                     batched_rows=batcher.rows_seen)
        return d

    @app.get("/metrics")
    def metrics():
        if prom is None:
            return JSONResponse({"error": "prometheus_client not installed"},
                                status_code=501)
        from fastapi.responses import Response
        from prometheus_client import generate_latest
        return Response(generate_latest(prom["registry"]),
                        media_type="text/plain; version=0.0.4")

    @app.get("/v1/models")
    def models():
        return {"object": "list",
                "data": [{"id": model_name, "object": "model"}]}

    @app.post("/v1/completions")
    def completions(body: Dict):
        prompt = body.get("prompt", "")
        cfg = _gen_config(body)
        ids = tokenizer.encode(prompt)
        if body.get("stream"):
            return _sse_stream(ids, cfg, lambda piece: {
                "object": "text_completion.chunk",
                "choices": [{"text": piece, "index": 0}]})
        toks = _decode(ids, cfg)
        return JSONResponse({
            "object": "text_completion",
            "model": model_name,
            "choices": [{"text": tokenizer.decode(toks), "index": 0,
                         "finish_reason": "stop"}],
            "usage": {"prompt_tokens": len(ids),
                      "completion_tokens": len(toks),
                      "total_tokens": len(ids) + len(toks)},
        })

    @app.post("/v1/chat/completions")
    def chat_completions(body: Dict):
        messages = body.get("messages", [])
        cfg = _gen_config(body)
        ids: List[int] = []
        for m in messages:
            ids.extend(tokenizer.encode_message(m.get("role", "user"),
                                                m.get("content", "")))
        # open the assistant turn so decoding continues it
        ids.extend(tokenizer.encode_message("assistant", "")[:-1])
        if body.get("stream"):
            return _sse_stream(ids, cfg, lambda piece: {
                "object": "chat.completion.chunk",
                "choices": [{"delta": {"content": piece}, "index": 0}]},
                endpoint="chat")
        toks = _decode(ids, cfg, endpoint="chat")
        return JSONResponse({
            "object": "chat.completion",
            "model": model_name,
            "choices": [{"message": {"role": "assistant",
                                     "content": tokenizer.decode(toks)},
                         "index": 0, "finish_reason": "stop"}],
            "usage": {"prompt_tokens": len(ids),
                      "completion_tokens": len(toks),
                      "total_tokens": len(ids) + len(toks)},
        })

    return app


def main(argv=None):
    import argparse

    from ..data.tokenizer import ConversationTokenizer
    from ..models import DeepSeekTransformer
    from .loader import (find_latest_checkpoint, infer_config_from_state_dict,
                         load_checkpoint_smart)

    ap = argparse.ArgumentParser(description="LuminaAI-AMD inference server")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--device", default=None)
    ap.add_argument("--quantize", default=None,
                    choices=["int8", "int4", "fp8"])
    ap.add_argument("--max-batch", type=int, default=8,
                    help="continuous-batching slot count")
    ap.add_argument("--max-len", type=int, default=2048,
                    help="serving context window (KV cache columns)")
    ap.add_argument("--serving", default="continuous",
                    choices=["continuous", "dynamic"])
    args = ap.parse_args(argv)

    device = torch.device(args.device) if args.device else (
        torch.device("cuda") if torch.cuda.is_available()
        else torch.device("cpu"))
    path = args.checkpoint or find_latest_checkpoint()
    if path is None:
        raise SystemExit("no checkpoint found; pass --checkpoint")
    payload = load_checkpoint_smart(path)
    sd = payload["model_state_dict"]
    cfg = infer_config_from_state_dict(sd)
    model = DeepSeekTransformer(cfg)
    model.load_state_dict(sd, strict=False)
    model = model.to(device).eval()
    if device.type == "cuda":
        model = model.to(torch.bfloat16)
    if args.quantize:
        from ..ops.quant import quantize_model
        n = quantize_model(model, mode=args.quantize)
        print(f"quantized {n} Linear layers to {args.quantize}")

    app = create_app(model, ConversationTokenizer(),
                     max_batch=args.max_batch, serving=args.serving,
                     max_len=args.max_len)
    import uvicorn
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
