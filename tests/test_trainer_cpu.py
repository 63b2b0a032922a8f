import pytest
import numpy as np
import torch

from gan_deeplearning4j_amd.config import GanConfig, preset
from gan_deeplearning4j_amd.data import (
    CSVRecordReader,
    RecordReaderDataSetIterator,
    write_synthetic_csv,
)
from gan_deeplearning4j_amd.models import build_dcgan, build_mlp_gan
from gan_deeplearning4j_amd.train import GanTrainer, ReferenceProtocolTrainer
from gan_deeplearning4j_amd.train.gan_trainer import latent_grid


def small_cfg():
    cfg = GanConfig()
    cfg.train.use_gpu = False
    cfg.data.batch_size_per_worker = 16
    cfg.data.batch_size_pred = 32
    cfg.train.num_iterations = 2
    return cfg


def test_latent_grid():
    z = latent_grid(10, 2)
    assert z.shape == (100, 2)
    assert z.min() == -1 and z.max() == 1


def test_fast_trainer_mlp():
    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    gen, dis = build_mlp_gan(cfg, hidden=32)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    x = torch.rand(8, cfg.data.num_features)
    out1 = tr.step(x)
    out2 = tr.step(x)
    assert np.isfinite(float(out1["loss_d"])) and np.isfinite(float(out2["loss_g"]))
    grid = tr.sample_grid(4)
    assert grid.shape[0] == 16


def test_ema_generator():
    # fp32 EMA of G's params: exact recursion check over two steps,
    # swap-in/swap-out restores the live weights bit-exactly
    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    cfg.train.ema_decay = 0.5
    gen, dis = build_mlp_gan(cfg, hidden=16)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    ema_expect = [p.detach().float().clone() for p in tr.gen.parameters()]
    x = torch.rand(8, cfg.data.num_features)
    for _ in range(2):
        tr.step(x)
        for e, p in zip(ema_expect, tr.gen.parameters()):
            e.mul_(0.5).add_(p.detach().float(), alpha=0.5)
    for e, a in zip(ema_expect, tr._ema):
        assert torch.allclose(e, a, atol=1e-7)
    live = [p.detach().clone() for p in tr.gen.parameters()]
    with tr.ema_weights():
        for p, e in zip(tr.gen.parameters(), tr._ema):
            assert torch.allclose(p.detach().float(), e, atol=1e-6)
    for p, b in zip(tr.gen.parameters(), live):
        assert torch.equal(p.detach(), b)
    # default off
    cfg2 = preset("mlp_tabular_cpu")
    cfg2.train.use_gpu = False
    gen2, dis2 = build_mlp_gan(cfg2, hidden=16)
    tr2 = GanTrainer(gen2, dis2, cfg2, device=torch.device("cpu"))
    assert tr2._ema is None
    with tr2.ema_weights() as g:
        assert g is tr2.gen


@pytest.mark.parametrize("loss_type", ["lsgan", "hinge"])
def test_alternative_adversarial_losses(loss_type):
    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    cfg.train.loss_type = loss_type
    gen, dis = build_mlp_gan(cfg, hidden=16)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    x = torch.rand(8, cfg.data.num_features)
    w0 = gen.params_flat().clone()
    out = None
    for _ in range(3):
        out = tr.step(x)
    assert np.isfinite(float(out["loss_d"]))
    assert np.isfinite(float(out["loss_g"]))
    # the generator actually trains under the alternative objective
    assert not torch.allclose(gen.params_flat(), w0)
    # unknown loss rejected
    cfg.train.loss_type = "wasserstein-gp"
    with pytest.raises(ValueError):
        GanTrainer(*build_mlp_gan(cfg, hidden=16), cfg,
                   device=torch.device("cpu"))


def test_fast_trainer_dcgan28_step():
    cfg = preset("dcgan28")
    cfg.train.use_gpu = False
    cfg.model.base_width = 8  # small for CPU test speed
    gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    real = torch.rand(4, 1, 28, 28) * 2 - 1
    out = tr.step(real)
    assert np.isfinite(float(out["loss_d"]))
    # G step must not have updated D
    # (D was stop-grad frozen during G step; its updater ran only once)
    assert tr.dis.updater.t == 1
    assert tr.gen.updater.t == 1


def test_reference_protocol_two_iterations(tmp_path):
    """Fixed-seed 2-iteration end-to-end replica of the reference run
    (numIterations=2, Java:72), asserting artifact shapes."""
    cfg = small_cfg()
    train_csv = write_synthetic_csv(tmp_path / "train.csv", "pixel_lattice",
                                    n=64, height=28, width=28, channels=1)
    test_csv = write_synthetic_csv(tmp_path / "test.csv", "pixel_lattice",
                                   n=32, height=28, width=28, channels=1)
    train_it = RecordReaderDataSetIterator(
        CSVRecordReader().initialize(train_csv),
        cfg.data.batch_size_per_worker, cfg.data.label_index,
        cfg.data.num_classes)
    test_it = RecordReaderDataSetIterator(
        CSVRecordReader().initialize(test_csv),
        cfg.data.batch_size_pred, cfg.data.label_index, cfg.data.num_classes)

    tr = ReferenceProtocolTrainer(cfg, device=torch.device("cpu"),
                                  out_dir=str(tmp_path / "out"))
    # init-time shape smoke (reference Java:167-170, 223-225, 312-314,
    # 365-368): D/CV map features->1/10 classes, G/GAN map z->image/1
    nf = cfg.data.num_features
    assert tr.smoke_shapes["dis"] == (10, 1)
    assert tr.smoke_shapes["cv"] == (10, cfg.data.num_classes)
    assert tr.smoke_shapes["gen"][0] == 10
    assert int(np.prod(tr.smoke_shapes["gen"][1:])) == nf
    assert tr.smoke_shapes["gan"] == (10, 1)
    last = tr.run(train_it, test_it)
    assert tr.batch_counter == 2
    assert np.isfinite(last["loss_d"]) and np.isfinite(last["loss_g"])

    # artifacts: 10x10 grid of 784-float rows (mnist_out_<i>.csv)
    grid = np.loadtxt(tmp_path / "out" / "mnist_out_2.csv", delimiter=",")
    assert grid.shape == (100, 784)
    preds = np.loadtxt(tmp_path / "out" / "mnist_test_predictions_2.csv",
                       delimiter=",")
    assert preds.shape == (32, 10)
    assert np.allclose(preds.sum(axis=1), 1.0, atol=1e-4)
    # 4 model zips
    for name in ("dis", "gan", "gen", "CV"):
        assert (tmp_path / "out" / f"mnist_{name}_model.zip").exists()


def test_reference_weight_sync_semantics(tmp_path):
    """After an iteration, gan's frozen D == dis, gen == gan's G."""
    cfg = small_cfg()
    cfg.train.train_classifier = False
    p = write_synthetic_csv(tmp_path / "t.csv", "pixel_lattice", n=32,
                            height=28, width=28, channels=1)
    it = RecordReaderDataSetIterator(CSVRecordReader().initialize(p),
                                     16, 784, 10)
    tr = ReferenceProtocolTrainer(cfg, device=torch.device("cpu"),
                                  out_dir=str(tmp_path / "o"))
    tr.train_iteration(next(iter(it)))
    assert torch.equal(
        tr.dis.get_layer("dis_conv2d_layer_2").get_param("W"),
        tr.gan.get_layer("gan_dis_conv2d_layer_10").get_param("W"),
    )
    assert torch.equal(
        tr.gen.get_layer("gen_dense_layer_2").get_param("W"),
        tr.gan.get_layer("gan_dense_layer_2").get_param("W"),
    )


def test_n_critic_d_steps():
    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    cfg.train.d_steps_per_g = 3
    gen, dis = build_mlp_gan(cfg, hidden=16)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    x = torch.rand(8, cfg.data.num_features)
    out = tr.step(x)
    assert tr.dis.updater.t == 3     # three D updates
    assert tr.gen.updater.t == 1     # one G update
    assert np.isfinite(float(out["loss_d"]))
    tr.step(x)
    assert tr.dis.updater.t == 6 and tr.gen.updater.t == 2
