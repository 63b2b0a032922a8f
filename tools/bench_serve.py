"""Serving load benchmark: concurrent generate requests through the
micro-batcher + compiled hipGraph replay.

    python tools/bench_serve.py [clients] [requests_per_client] [n_per_req] [arch]

Measures end-to-end requests/s and images/s for the serving path
(`serve._Endpoint` + `serve._MicroBatcher`), no HTTP framing — the
compute-side capacity number for docs/ROADMAP.md item 4.
"""

import asyncio
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.serve import _Endpoint, _MicroBatcher


def main():
    clients = int(sys.argv[1]) if len(sys.argv) > 1 else 32
    per_client = int(sys.argv[2]) if len(sys.argv) > 2 else 50
    n_per_req = int(sys.argv[3]) if len(sys.argv) > 3 else 4
    arch = sys.argv[4] if len(sys.argv) > 4 else "dcgan64"
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if dev.type == "cuda" else torch.float32
    cfg = preset(arch)
    gen, _ = build_dcgan(cfg)
    ep = _Endpoint(gen, dev, dtype, max_batch=clients * n_per_req)
    mb = _MicroBatcher(ep, window_ms=2.0)

    # warm the capture path once
    ep.run(torch.randn(2, cfg.model.z_size))
    if dev.type == "cuda":
        torch.cuda.synchronize()

    async def client(i):
        for _ in range(per_client):
            z = torch.randn(n_per_req, cfg.model.z_size)
            out = await mb.submit(z)
            assert out.shape[0] == n_per_req

    async def run():
        t0 = time.perf_counter()
        await asyncio.gather(*(client(i) for i in range(clients)))
        return time.perf_counter() - t0

    elapsed = asyncio.run(run())
    total_req = clients * per_client
    total_img = total_req * n_per_req
    print(f"{arch} serving: {clients} clients x {per_client} req x "
          f"{n_per_req} images: {total_req / elapsed:,.0f} req/s, "
          f"{total_img / elapsed:,.0f} images/s "
          f"({elapsed / total_req * 1e3:.3f} ms/req avg)")


if __name__ == "__main__":
    main()
