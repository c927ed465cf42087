"""HTTP serving front over the continuous batcher.

The reference repo references a flask ``server.py`` from its run script
and tests (/root/reference/cmd.sh:5-13, tests/test_server.py:14-22) but
the file is absent — the HTTP surface is dead there. This is the working
equivalent: a FastAPI app over `serving.ContinuousBatcher`, one
background decode thread driving `step()`, every concurrent request
sharing decode steps (continuous batching) instead of queueing whole
generations.

Endpoints:
  GET  /health            -> {"status": "ok", "pending": N}
  POST /generate          -> {"text": ..., "tokens": [...]}
      body: {"prompt": str, "num_tokens": int = 50,
             "temperature": float = 0.0 (0 = greedy),
             "repeat_penalty": float = 1.1, "seed": int | None,
             "top_k": int = 0, "top_p": float = 1.0,
             "stop": [str, ...] | None}  # end at first stop string
  POST /generate_stream   -> SSE, one data: line per token piece
  POST /v1/completions    -> OpenAI-compatible completions shape
  GET  /metrics           -> lifetime counters incl. tokens_per_step
"""
from __future__ import annotations

import threading
import time
from typing import Optional

from pydantic import BaseModel

from ..engine.sampler import Sampler


class GenerateRequest(BaseModel):
    prompt: str
    num_tokens: int = 50
    temperature: float = 0.0
    repeat_penalty: float = 1.1
    seed: Optional[int] = None
    top_k: int = 0
    top_p: float = 1.0
    stop: Optional[list] = None    # stop strings: generation ends when
    #                                the decoded continuation contains
    #                                one; the stop text is not returned


def trim_at_stop(tokenizer, ids, stops):
    """(ids, text) trimmed at the FIRST occurrence of any stop string,
    stop text excluded (OpenAI `stop` semantics). Token-granular: the
    cut keeps the shortest token prefix whose decoded text contains no
    stop — the returned text is cut mid-token-piece when a stop lands
    inside one."""
    text = tokenizer.decode(ids)
    cut = min((p for p in (text.find(s) for s in stops) if p >= 0),
              default=-1)
    if cut < 0:
        return list(ids), text, False
    keep = []
    for i, t in enumerate(ids):
        if len(tokenizer.decode(ids[:i + 1])) > cut:
            break
        keep.append(t)
    return keep, text[:cut], True


class BatcherWorker:
    """Drives ContinuousBatcher.step() on one thread; submissions from
    request handlers are lock-guarded (the batcher itself is not
    thread-safe). Each step broadcasts a condition so waiters can check
    their request without polling the GPU."""

    def __init__(self, batcher):
        self.batcher = batcher
        self.cond = threading.Condition()
        self._stop = False
        self.thread = threading.Thread(target=self._run, daemon=True)

    def start(self) -> "BatcherWorker":
        self.thread.start()
        return self

    def stop(self) -> None:
        with self.cond:
            self._stop = True
            self.cond.notify_all()
        self.thread.join(timeout=10)

    def _run(self) -> None:
        while True:
            with self.cond:
                if self._stop:
                    return
                if not self.batcher.pending:
                    self.cond.wait(timeout=0.05)
                    continue
                try:
                    self.batcher.step()
                except Exception:  # one bad step must not kill serving
                    import traceback
                    traceback.print_exc()
                    # fail every in-flight request rather than hang its
                    # waiter; slots are reclaimed through cancel()
                    for r in (list(self.batcher.active.values())
                              + list(self.batcher.prefilling.values())
                              + list(self.batcher.queue)):
                        self.batcher.cancel(r)
                self.cond.notify_all()

    def submit_and_wait(self, prompt_ids, max_new,
                        sampler: Optional[Sampler],
                        timeout: float = 300.0,
                        stop_check=None):
        """stop_check(req) -> bool: called on every step wake; True
        cancels the request (used for stop-string termination — the
        caller trims the output afterwards)."""
        with self.cond:
            req = self.batcher.submit(prompt_ids, max_new, sampler=sampler)
            self.cond.notify_all()
            deadline = time.monotonic() + timeout
            seen = 0
            while not req.done:
                if stop_check is not None and len(req.out) > seen:
                    seen = len(req.out)
                    if stop_check(req):
                        self.batcher.cancel(req)
                        break
                left = deadline - time.monotonic()
                if left <= 0:
                    # reclaim the KV slot — an abandoned request must
                    # not keep generating
                    self.batcher.cancel(req)
                    raise TimeoutError("generation timed out")
                self.cond.wait(timeout=min(left, 1.0))
        return req


def build_app(batcher, tokenizer, eos_id: Optional[int] = None):
    """FastAPI app + started worker; returns (app, worker)."""
    from fastapi import FastAPI, HTTPException

    app = FastAPI(title="distllm-mi355x")
    worker = BatcherWorker(batcher).start()
    stats = {"requests": 0, "tokens": 0, "t0": time.monotonic()}

    @app.get("/health")
    def health():
        return {"status": "ok", "pending": batcher.pending}

    @app.get("/metrics")
    def metrics():
        up = time.monotonic() - stats["t0"]
        return {"requests_served": stats["requests"],
                "tokens_generated": stats["tokens"],
                "pending": batcher.pending,
                "slots": batcher.n_slots,
                "lanes": len(batcher.lanes),
                "decode_steps": batcher.steps_run,
                "tokens_per_step": round(
                    batcher.tokens_out / max(batcher.steps_run, 1), 3),
                "uptime_s": round(up, 1),
                "tokens_per_s_lifetime": round(
                    stats["tokens"] / max(up, 1e-9), 1)}

    def _submit(r: GenerateRequest):
        if r.num_tokens < 1:
            raise HTTPException(422, "num_tokens must be >= 1")
        ids = tokenizer.encode(r.prompt, bos=True)
        sampler = None
        if r.temperature > 0.0:
            sampler = Sampler(r.temperature, r.repeat_penalty, seed=r.seed,
                              top_k=r.top_k, top_p=r.top_p)
        return ids, sampler

    @app.post("/generate")
    def generate(r: GenerateRequest):
        ids, sampler = _submit(r)
        stops = [s for s in (r.stop or []) if s]
        check = None
        if stops:
            def check(req):
                return any(s in tokenizer.decode(req.out) for s in stops)
        try:
            req = worker.submit_and_wait(ids, r.num_tokens, sampler,
                                         stop_check=check)
        except ValueError as e:        # oversized prompt+num_tokens
            raise HTTPException(422, str(e))
        except TimeoutError as e:
            raise HTTPException(504, str(e))
        toks, text = list(req.out), tokenizer.decode(req.out)
        if stops:
            toks, text, _ = trim_at_stop(tokenizer, req.out, stops)
        stats["requests"] += 1
        stats["tokens"] += len(toks)
        return {"text": text, "tokens": toks}

    @app.post("/generate_stream")
    def generate_stream(r: GenerateRequest):
        """Server-sent events: one `data:` line per decoded token as the
        batcher produces it, then a final `done` event with the full
        text — tokens stream while OTHER requests share the same decode
        steps."""
        import json as _json

        from fastapi.responses import StreamingResponse

        ids, sampler = _submit(r)
        stops = [s for s in (r.stop or []) if s]
        with worker.cond:
            try:
                req = batcher.submit(ids, r.num_tokens, sampler=sampler)
            except ValueError as e:
                raise HTTPException(422, str(e))
            worker.cond.notify_all()

        def events():
            sent = 0
            text = ""
            try:
                while True:
                    with worker.cond:
                        while len(req.out) == sent and not req.done:
                            worker.cond.wait(timeout=1.0)
                        chunk = list(req.out[sent:])
                        sent = len(req.out)
                        done = req.done
                    for tid in chunk:
                        piece = tokenizer.decode_token(tid)
                        if stops:
                            cand = text + piece
                            cut = min((p for p in (cand.find(x)
                                                   for x in stops)
                                       if p >= 0), default=-1)
                            if cut >= 0:
                                # stop found: emit only up to the cut,
                                # free the slot, finish the stream (a
                                # stop spanning pieces may have partly
                                # streamed — the done event carries the
                                # trimmed text)
                                tail = cand[len(text):cut]
                                if tail:
                                    yield ("data: " + _json.dumps(
                                        {"token": tid, "piece": tail})
                                        + "\n\n")
                                with worker.cond:
                                    batcher.cancel(req)
                                yield ("event: done\ndata: " +
                                       _json.dumps({"text": cand[:cut],
                                                    "stopped": True})
                                       + "\n\n")
                                return
                            text = cand
                        yield ("data: " + _json.dumps(
                            {"token": tid, "piece": piece}) + "\n\n")
                    if done:
                        yield ("event: done\ndata: " + _json.dumps(
                            {"text": tokenizer.decode(req.out),
                             "tokens": list(req.out)}) + "\n\n")
                        return
            finally:
                # client went away mid-stream (GeneratorExit) or any
                # failure: free the KV slot instead of decoding to
                # max_new for nobody
                if not req.done:
                    with worker.cond:
                        batcher.cancel(req)

        return StreamingResponse(events(), media_type="text/event-stream")

    @app.post("/v1/completions")
    def v1_completions(body: dict):
        """OpenAI-compatible completions endpoint (non-streaming): lets
        standard clients point at this server unchanged. Maps onto the
        same batcher path as /generate."""
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):        # the API allows a list
            if len(prompt) != 1:
                raise HTTPException(
                    400, "only a single prompt per request is supported")
            prompt = prompt[0]
        if not isinstance(prompt, str) or not prompt:
            raise HTTPException(400, "prompt must be a non-empty string")
        stop = body.get("stop")
        if isinstance(stop, str):
            stop = [stop]
        r = GenerateRequest(
            prompt=prompt,
            num_tokens=int(body.get("max_tokens", 16)),
            temperature=float(body.get("temperature", 0.0)),
            top_p=float(body.get("top_p", 1.0)),
            seed=body.get("seed"),
            stop=stop)
        out = generate(r)
        stats["v1_id"] = stats.get("v1_id", 0) + 1
        return {
            "id": f"cmpl-{stats['v1_id']}",
            "object": "text_completion",
            "model": body.get("model", "distllm-mi355x"),
            "choices": [{
                "text": out["text"],
                "index": 0,
                "logprobs": None,
                "finish_reason":
                    "length" if len(out["tokens"]) >= r.num_tokens
                    else "stop",
            }],
            "usage": {
                "prompt_tokens": len(tokenizer.encode(prompt, bos=True)),
                "completion_tokens": len(out["tokens"]),
                "total_tokens":
                    len(tokenizer.encode(prompt, bos=True)) +
                    len(out["tokens"]),
            },
        }

    app.state.worker = worker
    return app, worker
