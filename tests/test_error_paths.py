"""Error-path behavior: malformed graphs, bad config overrides, corrupt
checkpoints and missing-extension handling must fail loudly and clearly."""

import zipfile

import pytest
import torch

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.graph.builder import GraphBuilder, InputType
from gan_deeplearning4j_amd.graph.layers import DenseLayer


def _gb():
    gb = GraphBuilder(seed=1, optim_cfg=GanConfig().optim)
    gb.add_inputs("x")
    gb.set_input_types(InputType.feed_forward(4))
    return gb


def test_cycle_detection():
    gb = _gb()
    gb.add_layer("a", DenseLayer(4, 4), "b")
    gb.add_layer("b", DenseLayer(4, 4), "a")
    gb.set_outputs("b")
    with pytest.raises(ValueError, match="cycle"):
        gb.build()


def test_unknown_config_override():
    cfg = GanConfig()
    with pytest.raises(KeyError, match="unknown config key"):
        cfg.apply_overrides(["nope.key=1"])


def test_unknown_param_key():
    layer = DenseLayer(4, 4)
    with pytest.raises(KeyError):
        layer.get_param("gamma")


def test_corrupt_checkpoint_zip(tmp_path):
    from gan_deeplearning4j_amd.graph.serialization import ModelSerializer

    p = tmp_path / "bad.zip"
    p.write_bytes(b"this is not a zip")
    with pytest.raises(zipfile.BadZipFile):
        ModelSerializer.restore_computation_graph(p)


def test_truncated_coefficients(tmp_path):
    """A coefficients.bin with the wrong length must not load silently."""
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.graph.serialization import ModelSerializer
    from gan_deeplearning4j_amd.models import build_dcgan

    cfg = preset("dcgan28")
    gen, _ = build_dcgan(cfg)
    p = ModelSerializer.write_model(gen, tmp_path / "g.zip",
                                    save_updater=False)
    # rewrite the zip with a truncated coefficient blob
    with zipfile.ZipFile(p) as zf:
        conf = zf.read("configuration.json")
        coef = zf.read("coefficients.bin")
    bad = tmp_path / "bad.zip"
    with zipfile.ZipFile(bad, "w") as zf:
        zf.writestr("configuration.json", conf)
        zf.writestr("coefficients.bin", coef[: len(coef) // 2])
    with pytest.raises(Exception):
        ModelSerializer.restore_computation_graph(bad)


def test_gpu_ops_require_extension_or_gpu():
    """CPU tensors must never reach the GPU op layer's kernels; the
    functional API dispatches them to the torch path instead."""
    from gan_deeplearning4j_amd.ops import functional as OF

    x = torch.rand(2, 3, 8, 8)
    w = torch.rand(4, 3, 3, 3)
    y = OF.conv2d(x, w, None, 1, 1, "lrelu")
    assert y.shape == (2, 4, 8, 8)  # plain torch path


def test_serve_missing_model_404():
    fastapi = pytest.importorskip("fastapi")  # noqa: F841
    from fastapi.testclient import TestClient

    from gan_deeplearning4j_amd.serve import create_app

    app = create_app(device=torch.device("cpu"))
    c = TestClient(app)
    assert c.post("/generate", json={"n": 1}).status_code == 404
    assert c.post("/discriminate", json={"inputs": [[0.0]]}).status_code == 404
