"""REST API server.

Endpoint-for-endpoint port of the reference's FastAPI façade
(``tensorlink/api/node.py:94-541``): /v1/generate, /v1/chat/completions,
/request-model, /model-status, /model-demand, /models, /stats,
/network-history, /node-info — but the request path is a direct in-process
call into :class:`InferenceEngine` instead of the reference's
cross-process endpoint_requests polling (``api/node.py:395-455``,
``ml/validator.py:534-556``). Streaming responses run the decode loop in a
worker thread and yield SSE chunks straight from the token queue.

Keeps the reference's per-IP rate limiting concept
(``tensorlink/p2p/monitor.py``) for the REST surface only.
"""

from __future__ import annotations

import threading
import time
from collections import defaultdict, deque
from typing import Optional

from fastapi import FastAPI, HTTPException, Request
from fastapi.responses import JSONResponse, StreamingResponse

from tensorlink_amd.api.schemas import (ChatCompletionRequest,
                                        GenerationRequest, JobRequest)
from tensorlink_amd.engine.engine import InferenceEngine


class RateLimiter:
    """Sliding-window per-IP limiter (reference p2p/monitor.py:4-43:
    5 attempts/min, 600 s block — applied here to job requests)."""

    def __init__(self, max_per_minute: int = 30, block_s: int = 600):
        self.max_per_minute = max_per_minute
        self.block_s = block_s
        self.hits = defaultdict(deque)
        self.blocked = {}
        self._lock = threading.Lock()

    def allow(self, ip: str) -> bool:
        now = time.time()
        with self._lock:
            if ip in self.blocked:
                if now < self.blocked[ip]:
                    return False
                del self.blocked[ip]
            dq = self.hits[ip]
            while dq and dq[0] < now - 60:
                dq.popleft()
            dq.append(now)
            if len(dq) > self.max_per_minute:
                self.blocked[ip] = now + self.block_s
                return False
            return True


def create_app(engine: InferenceEngine,
               node_info: Optional[dict] = None) -> FastAPI:
    app = FastAPI(title="tensorlink-amd", version="0.1.0")
    limiter = RateLimiter()
    started = time.time()

    def _check_rate(request: Request):
        ip = request.client.host if request.client else "unknown"
        if not limiter.allow(ip):
            raise HTTPException(429, "rate limited")

    # ---------------- generation ----------------
    @app.post("/v1/generate")
    def generate(req: GenerationRequest, request: Request):
        payload = req.model_dump()
        if req.stream:
            return StreamingResponse(engine.generate_stream(payload),
                                     media_type="text/event-stream")
        result = engine.generate(payload)
        if "error" in result:
            return JSONResponse(result, status_code=result["error"].get(
                "code", 500))
        return result

    @app.post("/v1/chat/completions")
    def chat_completions(req: ChatCompletionRequest, request: Request):
        # map OpenAI chat request onto a GenerationRequest (reference
        # api/node.py:198-210)
        history = [m.model_dump() for m in req.messages]
        message = ""
        if history and history[-1]["role"] == "user":
            message = history.pop()["content"]
        payload = {
            "hf_name": req.model,
            "message": message,
            "history": history,
            "max_new_tokens": req.max_completion_tokens or req.max_tokens,
            "temperature": req.temperature,
            "top_p": req.top_p,
            "do_sample": req.temperature > 0,
            "stream": req.stream,
            "stop": req.stop,
            "n": getattr(req, "n", 1) or 1,
            "presence_penalty": getattr(req, "presence_penalty", 0.0) or 0.0,
            "frequency_penalty": getattr(req, "frequency_penalty", 0.0)
            or 0.0,
            "input_format": "chat",
            "output_format": "openai",
        }
        if req.stream:
            return StreamingResponse(engine.generate_stream(payload),
                                     media_type="text/event-stream")
        result = engine.generate(payload)
        if "error" in result:
            return JSONResponse(result, status_code=500)
        return result

    # ---------------- model management ----------------
    @app.post("/request-model")
    def request_model(req: JobRequest, request: Request):
        _check_rate(request)
        try:
            job = engine.load_model(
                req.hf_name, continuous=req.continuous,
                max_slots=req.max_slots, max_ctx=req.max_ctx,
                prefill_chunk=req.prefill_chunk,
                prefix_caching=req.prefix_caching,
                quantize=req.quantize)
            return {"model": req.hf_name, "status": job.state,
                    "continuous": job.batcher is not None}
        except KeyError as e:
            raise HTTPException(404, str(e))
        except Exception as e:
            raise HTTPException(500, str(e))

    @app.get("/model-status")
    def model_status(hf_name: str):
        return engine.model_status(hf_name)

    @app.get("/model-demand")
    def model_demand():
        return engine.model_demand()

    @app.get("/models")
    def models():
        return {"models": engine.models()}

    # ---------------- node status ----------------
    @app.get("/stats")
    def stats():
        return engine.stats()

    @app.get("/network-history")
    def network_history():
        return {"history": engine.metrics.snapshot()["history"]}

    @app.get("/health")
    def health():
        from tensorlink_amd.utils.watchdog import check_gpu_health
        h = check_gpu_health()
        code = 200 if h["status"] != "failed" else 503
        return JSONResponse({"status": h["status"],
                             "checks": h["checks"],
                             "models": len(engine.jobs)},
                            status_code=code)

    @app.get("/metrics")
    def prometheus_metrics():
        # Prometheus exposition (beyond the reference's JSON /stats —
        # production serving needs a scrapeable endpoint)
        from prometheus_client import (CONTENT_TYPE_LATEST,
                                       generate_latest)
        from starlette.responses import Response
        _sync_prometheus(engine)
        return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

    @app.get("/proposal-history")
    def proposal_history():
        # reference api/node.py:336 serves on-chain governance
        # proposals; the chain layer is deleted by design on a single
        # node, so the endpoint exists with an empty history
        return {"proposals": [], "note": "no on-chain governance in "
                "the single-node build"}

    @app.get("/node-info")
    def node_info_ep():
        info = {"version": "0.1.0", "role": "engine",
                "world_size": engine.world,
                "uptime_s": time.time() - started}
        if node_info:
            info.update(node_info)
        return info

    return app


_PROM = {}


def _sync_prometheus(engine):
    """Mirror EngineMetrics counters into prometheus_client gauges
    (created lazily so importing the server never requires a scrape)."""
    from prometheus_client import Gauge
    if not _PROM:
        _PROM["requests"] = Gauge("tl_requests_total",
                                  "requests served")
        _PROM["errors"] = Gauge("tl_errors_total", "failed requests")
        _PROM["tokens"] = Gauge("tl_tokens_total", "tokens generated")
        _PROM["uptime"] = Gauge("tl_uptime_seconds", "engine uptime")
        _PROM["latency_p50"] = Gauge("tl_latency_p50_seconds",
                                     "request latency p50")
        _PROM["models"] = Gauge("tl_models_loaded", "loaded models")
    m = engine.metrics.snapshot()
    _PROM["requests"].set(m.get("requests_total", 0))
    _PROM["errors"].set(m.get("errors_total", 0))
    _PROM["tokens"].set(m.get("tokens_total", 0))
    _PROM["uptime"].set(m.get("uptime_s", 0))
    _PROM["latency_p50"].set(m.get("latency_p50_s") or 0)
    _PROM["models"].set(len(engine.jobs))


class TensorlinkAPI:
    """uvicorn wrapper matching the reference's threaded server
    (``api/node.py:523-541``)."""

    def __init__(self, engine: InferenceEngine, host: str = "127.0.0.1",
                 port: int = 8000):
        self.engine = engine
        self.app = create_app(engine)
        self.host = host
        self.port = port
        self._server = None
        self._thread = None

    def start(self, background: bool = True):
        import uvicorn
        config = uvicorn.Config(self.app, host=self.host, port=self.port,
                                log_level="warning")
        self._server = uvicorn.Server(config)
        if background:
            self._thread = threading.Thread(target=self._server.run,
                                            daemon=True)
            self._thread.start()
        else:
            self._server.run()

    def stop(self):
        if self._server is not None:
            self._server.should_exit = True
