import zipfile

import numpy as np
import torch

from gan_deeplearning4j_amd.config import GanConfig, preset
from gan_deeplearning4j_amd.data.csv_reader import DataSet
from gan_deeplearning4j_amd.graph.serialization import ModelSerializer
from gan_deeplearning4j_amd.models import build_discriminator, build_dcgan


def test_zip_layout(tmp_path):
    # DL4J ModelSerializer zip layout:
    # configuration.json + coefficients.bin (+ updaterState.bin)
    dis = build_discriminator(GanConfig())
    dis.fit(DataSet(torch.rand(8, 784), torch.ones(8, 1)))  # create updater
    p = ModelSerializer.write_model(dis, tmp_path / "m.zip", save_updater=True)
    with zipfile.ZipFile(p) as zf:
        names = set(zf.namelist())
        assert "configuration.json" in names
        assert "coefficients.bin" in names
        assert "updaterState.bin" in names
        coef = np.frombuffer(zf.read("coefficients.bin"), dtype="<f4")
    assert coef.size == dis.params_flat().numel()
    # coefficients are the flattened params in layer order
    assert np.allclose(coef, dis.params_flat().numpy())


def test_checkpoint_roundtrip(tmp_path):
    dis = build_discriminator(GanConfig())
    ds = DataSet(torch.rand(8, 784), torch.ones(8, 1))
    dis.fit(ds)
    p = ModelSerializer.write_model(dis, tmp_path / "m.zip")
    back = ModelSerializer.restore_computation_graph(p)
    x = torch.rand(4, 784)
    assert torch.allclose(dis.output(x), back.output(x), atol=1e-6)
    # resume training continues identically (updater state restored)
    torch.manual_seed(0)
    l1 = dis.fit(ds)
    torch.manual_seed(0)
    l2 = back.fit(ds)
    assert abs(l1 - l2) < 1e-6


def test_native_roundtrip(tmp_path):
    gen, dis = build_dcgan(preset("dcgan28"))
    p = ModelSerializer.save_native(gen, tmp_path / "g.pt")
    back = ModelSerializer.load_native(p)
    z = torch.randn(4, 2)
    assert torch.allclose(gen.output(z), back.output(z), atol=1e-6)
