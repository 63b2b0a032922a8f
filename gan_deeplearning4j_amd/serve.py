"""HTTP serving for trained GAN graphs (hipGraph-compiled inference).

The reference repo has no serving story (models end at offline CSV dumps,
`gan.ipynb` cell 6); this module closes the production loop for SURVEY.md
§5 "serving/inference": a FastAPI app exposing generator sampling and
discriminator scoring, with the compute path captured once as a hipGraph
(`ComputationGraph.compile_inference`) and replayed per request.

Serving idiom for MI355X: requests are padded to ONE fixed batch shape so
a single captured graph serves every request (capture is per-shape); the
pad rows ride along and are sliced off. On CPU (tests, dev) the same API
runs the eager path.

    python -m gan_deeplearning4j_amd.serve --generator gen.zip \
        --discriminator dis.zip --port 8000
"""

from __future__ import annotations

import argparse
import base64
import tempfile
import threading
import time
from pathlib import Path
from typing import Optional

import torch

from .graph.builder import ComputationGraph
from .graph.serialization import ModelSerializer

try:  # web deps are optional for the core framework
    from pydantic import BaseModel, Field

    class GenerateReq(BaseModel):
        n: int = Field(default=1, ge=1)
        seed: Optional[int] = None
        format: str = Field(default="array", pattern="^(array|png_base64)$")
        model: Optional[str] = None  # named generator (default: "default")

    class DiscriminateReq(BaseModel):
        inputs: list
except ImportError:  # pragma: no cover
    GenerateReq = DiscriminateReq = None


def load_graph(path: str | Path) -> ComputationGraph:
    """Load a checkpoint in either supported format (DL4J zip / native)."""
    p = Path(path)
    if p.suffix == ".zip":
        return ModelSerializer.restore_computation_graph(p, load_updater=False)
    return ModelSerializer.load_native(p)


class _Endpoint:
    """One model + one captured fixed-shape inference graph."""

    def __init__(self, graph: ComputationGraph, device: torch.device,
                 dtype: torch.dtype, max_batch: int):
        self.graph = graph.to_device(device, dtype)
        self.graph.eval()
        self.device = device
        self.dtype = dtype
        self.max_batch = max_batch
        self._compiled = None
        self._example: Optional[torch.Tensor] = None
        # /discriminate is a sync route on FastAPI's thread pool: without
        # this lock concurrent requests race on the captured graph's
        # static input/output buffers (the /generate path is already
        # serialized by its micro-batcher)
        self._lock = threading.Lock()

    def _runner(self, example: torch.Tensor):
        if self._compiled is None or self._example.shape != example.shape:
            self._example = example
            self._compiled = self.graph.compile_inference(example)
        return self._compiled

    @torch.no_grad()
    def run(self, x: torch.Tensor) -> torch.Tensor:
        n = x.shape[0]
        if n > self.max_batch:
            raise ValueError(f"batch {n} > max_batch {self.max_batch}")
        with self._lock:
            # pad to the one captured shape
            xp = x.new_zeros(self.max_batch, *x.shape[1:])
            xp[:n] = x
            xp = xp.to(self.device, self.dtype)
            out = self._runner(xp)(xp)
            # .cpu() materializes a private copy before the lock releases
            return out[:n].float().cpu()


class _MicroBatcher:
    """Accumulates concurrent requests into one fixed-shape graph replay.

    Requests arriving within `window_ms` of each other (up to the
    endpoint's max_batch rows) share a single padded replay — the
    serving idiom for a captured hipGraph, where launch count, not
    per-row work, is the cost at small batches.
    """

    def __init__(self, endpoint: _Endpoint, window_ms: float = 2.0):
        self.ep = endpoint
        self.window = window_ms / 1000.0
        self.queue = None
        self._task = None
        self._loop = None

    def ensure_running(self):
        import asyncio

        loop = asyncio.get_running_loop()
        if self._loop is not loop:
            # fresh event loop (first request, or a per-request test
            # client): rebind the queue and worker to it
            self.queue = asyncio.Queue()
            self._loop = loop
            self._task = loop.create_task(self._run())
        elif self._task is None or self._task.done():
            self._task = loop.create_task(self._run())

    async def submit(self, x: torch.Tensor) -> torch.Tensor:
        import asyncio

        self.ensure_running()
        fut = asyncio.get_running_loop().create_future()
        await self.queue.put((x, fut))
        return await fut

    async def _run(self):
        import asyncio

        loop = asyncio.get_running_loop()
        while True:
            x, fut = await self.queue.get()
            items = [(x, fut)]
            rows = x.shape[0]
            deadline = loop.time() + self.window
            while rows < self.ep.max_batch:
                timeout = deadline - loop.time()
                if timeout <= 0:
                    break
                try:
                    nx, nfut = await asyncio.wait_for(self.queue.get(),
                                                      timeout)
                except asyncio.TimeoutError:
                    break
                if rows + nx.shape[0] > self.ep.max_batch:
                    # flush current batch; start the next with this item
                    self._dispatch(items)
                    items, rows = [], 0
                    deadline = loop.time() + self.window
                items.append((nx, nfut))
                rows += nx.shape[0]
            self._dispatch(items)

    def _dispatch(self, items):
        if not items:
            return
        xs = torch.cat([x for x, _ in items], dim=0)
        try:
            out = self.ep.run(xs)
        except Exception as e:  # propagate to every waiter
            for _, fut in items:
                if not fut.done():
                    fut.set_exception(e)
            return
        off = 0
        for x, fut in items:
            n = x.shape[0]
            if not fut.done():
                fut.set_result(out[off:off + n])
            off += n


def create_app(generator: Optional[ComputationGraph] = None,
               discriminator: Optional[ComputationGraph] = None,
               device: Optional[torch.device] = None,
               max_batch: int = 64,
               batch_window_ms: float = 2.0,
               generators: Optional[dict] = None):
    """Build the FastAPI app. Models may be None (their routes 404).

    generators: optional {name: ComputationGraph} registry of ADDITIONAL
    generators (multi-model residency — 288 GB HBM holds thousands of
    GAN-scale models; each gets its own captured graph + micro-batcher).
    `generator` is registered as "default".
    """
    from fastapi import FastAPI, HTTPException

    if device is None:
        device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

    gens: dict = {}
    if generator is not None:
        gens["default"] = generator
    for name, g in (generators or {}).items():
        gens[name] = g
    gen_eps = {name: _Endpoint(g, device, dtype, max_batch)
               for name, g in gens.items()}
    gen_mbs = {name: _MicroBatcher(ep, batch_window_ms)
               for name, ep in gen_eps.items()}
    dis_ep = (_Endpoint(discriminator, device, dtype, max_batch)
              if discriminator is not None else None)

    app = FastAPI(title="gan_deeplearning4j_amd serving")

    # Prometheus metrics (per-app registry so tests can build many apps)
    try:
        from prometheus_client import (CONTENT_TYPE_LATEST,
                                       CollectorRegistry, Counter,
                                       Histogram, generate_latest)

        _reg = CollectorRegistry()
        _req_count = Counter("serve_requests_total", "requests served",
                             ["endpoint", "model"], registry=_reg)
        _req_err = Counter("serve_errors_total", "request errors",
                           ["endpoint"], registry=_reg)
        _req_lat = Histogram("serve_request_seconds", "request latency",
                             ["endpoint"], registry=_reg)

        @app.get("/metrics")
        def metrics():
            from fastapi import Response

            return Response(generate_latest(_reg),
                            media_type=CONTENT_TYPE_LATEST)
    except ImportError:  # pragma: no cover - metrics become no-ops
        class _Null:
            def labels(self, *a, **k):
                return self

            def inc(self, *a):
                pass

            def observe(self, *a):
                pass

        _req_count = _req_err = _req_lat = _Null()

    @app.get("/healthz")
    def healthz():
        return {
            "status": "ok",
            "device": str(device),
            "generator": len(gen_eps) > 0,
            "discriminator": dis_ep is not None,
        }

    @app.get("/models")
    def models():
        return {"generators": sorted(gen_eps.keys()),
                "discriminator": dis_ep is not None}

    @app.get("/info")
    def info():
        out = {}
        for name, ep in gen_eps.items():
            key = "generator" if name == "default" else f"generator:{name}"
            out[key] = {"n_params": ep.graph.n_params(),
                        "inputs": ep.graph.input_names}
        if dis_ep is not None:
            out["discriminator"] = {"n_params": dis_ep.graph.n_params(),
                                    "inputs": dis_ep.graph.input_names}
        return out

    @app.post("/generate")
    async def generate(req: GenerateReq):
        name = req.model or "default"
        gen_ep = gen_eps.get(name)
        if gen_ep is None:
            _req_err.labels("generate").inc()
            raise HTTPException(404, f"no generator {name!r} loaded")
        gen_mb = gen_mbs[name]
        if req.n > max_batch:
            _req_err.labels("generate").inc()
            raise HTTPException(400, f"n > max_batch ({max_batch})")
        _req_count.labels("generate", name).inc()
        _t0 = time.monotonic()
        it = gen_ep.graph.input_types[gen_ep.graph.input_names[0]]
        z_size = it.shape(1)[1]
        g = torch.Generator().manual_seed(req.seed) if req.seed is not None \
            else None
        z = torch.randn(req.n, z_size, generator=g)
        # concurrent requests coalesce into one padded graph replay
        samples = await gen_mb.submit(z)
        _req_lat.labels("generate").observe(time.monotonic() - _t0)
        if req.format == "png_base64":
            from .utils.imaging import save_image_grid

            if samples.ndim == 2:  # flat image rows -> square grayscale
                side = int(samples.shape[1] ** 0.5)
                samples = samples.reshape(req.n, 1, side, side)
            with tempfile.TemporaryDirectory() as td:
                p = save_image_grid(samples, Path(td) / "grid.png",
                                    nrow=min(req.n, 10))
                data = p.read_bytes()
            return {"png_base64": base64.b64encode(data).decode()}
        return {"samples": samples.tolist()}

    @app.post("/discriminate")
    def discriminate(req: DiscriminateReq):
        if dis_ep is None:
            _req_err.labels("discriminate").inc()
            raise HTTPException(404, "no discriminator loaded")
        _req_count.labels("discriminate", "default").inc()
        x = torch.tensor(req.inputs, dtype=torch.float32)
        if x.ndim == 1:
            x = x.unsqueeze(0)
        if x.shape[0] > max_batch:
            raise HTTPException(400, f"batch > max_batch ({max_batch})")
        scores = dis_ep.run(x)
        return {"scores": scores.squeeze(-1).tolist()}

    return app


def main(argv: Optional[list[str]] = None):
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--generator", type=str, default=None)
    ap.add_argument("--discriminator", type=str, default=None)
    ap.add_argument("--add-generator", action="append", default=[],
                    metavar="NAME=PATH",
                    help="register an additional named generator")
    ap.add_argument("--host", type=str, default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch", type=int, default=64)
    args = ap.parse_args(argv)
    if args.generator is None and args.discriminator is None \
            and not args.add_generator:
        ap.error("load at least one model (--generator / --discriminator "
                 "/ --add-generator)")
    extra = {}
    for spec in args.add_generator:
        name, _, path = spec.partition("=")
        if not name or not path:
            ap.error(f"--add-generator wants NAME=PATH, got {spec!r}")
        extra[name] = load_graph(path)
    import uvicorn

    app = create_app(
        generator=load_graph(args.generator) if args.generator else None,
        discriminator=(load_graph(args.discriminator)
                       if args.discriminator else None),
        max_batch=args.max_batch,
        generators=extra,
    )
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
