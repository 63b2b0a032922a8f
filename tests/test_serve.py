"""Serving API tests (CPU eager path; on GPU the same app replays a
captured hipGraph — covered by tests/test_gpu_training.py's
compile_inference test)."""

import base64

import pytest
import torch

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from gan_deeplearning4j_amd.config import preset
from gan_deeplearning4j_amd.models import build_dcgan
from gan_deeplearning4j_amd.serve import create_app, load_graph


@pytest.fixture(scope="module")
def client():
    cfg = preset("dcgan28")
    gen, dis = build_dcgan(cfg)
    app = create_app(generator=gen, discriminator=dis,
                     device=torch.device("cpu"), max_batch=8)
    return TestClient(app)


def test_healthz_and_info(client):
    r = client.get("/healthz")
    assert r.status_code == 200
    assert r.json()["generator"] and r.json()["discriminator"]
    info = client.get("/info").json()
    assert info["generator"]["n_params"] > 0
    assert info["discriminator"]["inputs"] == ["d_input"]


def test_generate_array(client):
    r = client.post("/generate", json={"n": 3, "seed": 1})
    assert r.status_code == 200
    s = torch.tensor(r.json()["samples"])
    assert s.shape[0] == 3 and s.ndim == 4
    # seeded requests are reproducible
    r2 = client.post("/generate", json={"n": 3, "seed": 1})
    assert torch.allclose(s, torch.tensor(r2.json()["samples"]), atol=1e-5)
    r3 = client.post("/generate", json={"n": 3, "seed": 2})
    assert not torch.allclose(s, torch.tensor(r3.json()["samples"]))


def test_generate_png(client):
    r = client.post("/generate", json={"n": 4, "format": "png_base64"})
    assert r.status_code == 200
    png = base64.b64decode(r.json()["png_base64"])
    assert png[:8] == b"\x89PNG\r\n\x1a\n"


def test_generate_over_max_batch(client):
    r = client.post("/generate", json={"n": 99})
    assert r.status_code == 400


def test_discriminate(client):
    x = torch.rand(2, 784).tolist()
    r = client.post("/discriminate", json={"inputs": x})
    assert r.status_code == 200
    scores = r.json()["scores"]
    assert len(scores) == 2
    assert all(0.0 <= v <= 1.0 for v in scores)


def test_multi_model_registry():
    """Named generators serve side by side (multi-model residency)."""
    from gan_deeplearning4j_amd.serve import create_app as mk

    cfg28 = preset("dcgan28")
    g28, _ = build_dcgan(cfg28)
    cfg64 = preset("dcgan64")
    g64, _ = build_dcgan(cfg64)
    app = mk(generator=g28, generators={"hires": g64},
             device=torch.device("cpu"), max_batch=4)
    c = TestClient(app)
    assert c.get("/models").json()["generators"] == ["default", "hires"]
    s28 = torch.tensor(c.post("/generate", json={"n": 2}).json()["samples"])
    assert s28.shape == (2, 1, 28, 28)
    s64 = torch.tensor(c.post("/generate", json={
        "n": 2, "model": "hires"}).json()["samples"])
    assert s64.shape == (2, 3, 64, 64)
    assert c.post("/generate", json={"n": 1, "model": "nope"}
                  ).status_code == 404


def test_microbatcher_coalesces_concurrent_requests():
    """Concurrent submits within the window share ONE padded replay."""
    import asyncio

    from gan_deeplearning4j_amd.serve import _Endpoint, _MicroBatcher

    cfg = preset("dcgan28")
    gen, _ = build_dcgan(cfg)
    ep = _Endpoint(gen, torch.device("cpu"), torch.float32, max_batch=16)
    calls = []
    real_run = ep.run
    ep.run = lambda x: (calls.append(x.shape[0]), real_run(x))[1]
    mb = _MicroBatcher(ep, window_ms=50.0)

    async def go():
        zs = [torch.randn(2, cfg.model.z_size) for _ in range(4)]
        outs = await asyncio.gather(*(mb.submit(z) for z in zs))
        return zs, outs

    zs, outs = asyncio.run(go())
    assert calls == [8], calls  # 4 x 2 rows coalesced into one run
    for z, o in zip(zs, outs):
        assert torch.allclose(o, gen.output(z), atol=1e-5)


def test_microbatcher_splits_over_max_batch():
    """A flood larger than max_batch splits into multiple replays, each
    request still answered with its own rows."""
    import asyncio

    from gan_deeplearning4j_amd.serve import _Endpoint, _MicroBatcher

    cfg = preset("dcgan28")
    gen, _ = build_dcgan(cfg)
    ep = _Endpoint(gen, torch.device("cpu"), torch.float32, max_batch=4)
    calls = []
    real_run = ep.run
    ep.run = lambda x: (calls.append(x.shape[0]), real_run(x))[1]
    mb = _MicroBatcher(ep, window_ms=50.0)

    async def go():
        zs = [torch.randn(3, cfg.model.z_size) for _ in range(3)]
        return zs, await asyncio.gather(*(mb.submit(z) for z in zs))

    zs, outs = asyncio.run(go())
    assert sum(calls) == 9 and all(c <= 4 for c in calls), calls
    for z, o in zip(zs, outs):
        assert torch.allclose(o, gen.output(z), atol=1e-5)


@pytest.mark.gpu
def test_serve_gpu_compiled_path():
    """On MI355X the endpoint must serve through the captured hipGraph
    (bf16 compute) and padded fixed-shape batches."""
    cfg = preset("dcgan28")
    gen, dis = build_dcgan(cfg)
    app = create_app(generator=gen, discriminator=dis,
                     device=torch.device("cuda:0"), max_batch=16)
    c = TestClient(app)
    r = c.post("/generate", json={"n": 5, "seed": 3})
    assert r.status_code == 200
    s = torch.tensor(r.json()["samples"])
    assert s.shape == (5, 1, 28, 28) and torch.isfinite(s).all()
    # same seed -> same samples through graph replays
    r2 = c.post("/generate", json={"n": 5, "seed": 3})
    assert torch.allclose(s, torch.tensor(r2.json()["samples"]), atol=1e-2)
    x = torch.rand(3, 784).tolist()
    scores = c.post("/discriminate", json={"inputs": x}).json()["scores"]
    assert len(scores) == 3


def test_load_graph_roundtrip(tmp_path):
    from gan_deeplearning4j_amd.graph.serialization import ModelSerializer

    cfg = preset("dcgan28")
    gen, _ = build_dcgan(cfg)
    p = ModelSerializer.write_model(gen, tmp_path / "gen.zip")
    g2 = load_graph(p)
    z = torch.rand(2, cfg.model.z_size)
    assert torch.allclose(gen.output(z), g2.output(z), atol=1e-6)


def test_discriminate_concurrent_consistency(client):
    """Concurrent /discriminate requests must not race on the endpoint's
    static buffers (the _Endpoint.run lock, round-1 advisor finding):
    every thread's scores must equal the single-thread result for its
    own input."""
    import concurrent.futures

    torch.manual_seed(5)
    inputs = [torch.rand(2, 784).tolist() for _ in range(8)]
    want = [client.post("/discriminate", json={"inputs": x}).json()["scores"]
            for x in inputs]

    def hit(i):
        r = client.post("/discriminate", json={"inputs": inputs[i]})
        assert r.status_code == 200
        return i, r.json()["scores"]

    with concurrent.futures.ThreadPoolExecutor(max_workers=8) as ex:
        for i, scores in ex.map(hit, list(range(8)) * 4):
            assert scores == pytest.approx(want[i], abs=1e-5), i


def test_metrics_endpoint(client):
    # prometheus metrics: counters move with requests
    r = client.post("/generate", json={"n": 2})
    assert r.status_code == 200
    m = client.get("/metrics")
    assert m.status_code == 200
    body = m.text
    assert 'serve_requests_total{endpoint="generate",model="default"}' in body
    assert "serve_request_seconds" in body
    # errors counted
    client.post("/generate", json={"n": 10_000})
    body2 = client.get("/metrics").text
    assert 'serve_errors_total{endpoint="generate"}' in body2
