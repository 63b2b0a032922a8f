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
        # pad to the one captured shape
        xp = x.new_zeros(self.max_batch, *x.shape[1:])
        xp[:n] = x
        xp = xp.to(self.device, self.dtype)
        out = self._runner(xp)(xp)
        return out[:n].float().cpu()


def create_app(generator: Optional[ComputationGraph] = None,
               discriminator: Optional[ComputationGraph] = None,
               device: Optional[torch.device] = None,
               max_batch: int = 64):
    """Build the FastAPI app. Models may be None (their routes 404)."""
    from fastapi import FastAPI, HTTPException

    if device is None:
        device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32

    gen_ep = (_Endpoint(generator, device, dtype, max_batch)
              if generator is not None else None)
    dis_ep = (_Endpoint(discriminator, device, dtype, max_batch)
              if discriminator is not None else None)

    app = FastAPI(title="gan_deeplearning4j_amd serving")

    @app.get("/healthz")
    def healthz():
        return {
            "status": "ok",
            "device": str(device),
            "generator": gen_ep is not None,
            "discriminator": dis_ep is not None,
        }

    @app.get("/info")
    def info():
        out = {}
        if gen_ep is not None:
            out["generator"] = {"n_params": gen_ep.graph.n_params(),
                                "inputs": gen_ep.graph.input_names}
        if dis_ep is not None:
            out["discriminator"] = {"n_params": dis_ep.graph.n_params(),
                                    "inputs": dis_ep.graph.input_names}
        return out

    @app.post("/generate")
    def generate(req: GenerateReq):
        if gen_ep is None:
            raise HTTPException(404, "no generator loaded")
        if req.n > max_batch:
            raise HTTPException(400, f"n > max_batch ({max_batch})")
        it = gen_ep.graph.input_types[gen_ep.graph.input_names[0]]
        z_size = it.shape(1)[1]
        g = torch.Generator().manual_seed(req.seed) if req.seed is not None \
            else None
        z = torch.randn(req.n, z_size, generator=g)
        samples = gen_ep.run(z)
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
            raise HTTPException(404, "no discriminator loaded")
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
    ap.add_argument("--host", type=str, default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch", type=int, default=64)
    args = ap.parse_args(argv)
    if args.generator is None and args.discriminator is None:
        ap.error("load at least one of --generator/--discriminator")
    import uvicorn

    app = create_app(
        generator=load_graph(args.generator) if args.generator else None,
        discriminator=(load_graph(args.discriminator)
                       if args.discriminator else None),
        max_batch=args.max_batch,
    )
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
