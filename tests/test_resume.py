import torch

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.data.csv_reader import DataSet
from gan_deeplearning4j_amd.train import ReferenceProtocolTrainer


def test_reference_trainer_resume_roundtrip(tmp_path):
    cfg = GanConfig()
    cfg.train.use_gpu = False
    cfg.data.batch_size_per_worker = 16
    tr = ReferenceProtocolTrainer(cfg, device=torch.device("cpu"),
                                  out_dir=str(tmp_path))
    feats = torch.rand(16, 784)
    labels = torch.eye(10)[torch.randint(0, 10, (16,))]
    tr.train_iteration(DataSet(feats, labels))
    tr.save_checkpoints()
    ref_out = tr.gen.output(torch.rand(4, 2))

    tr2 = ReferenceProtocolTrainer(cfg, device=torch.device("cpu"),
                                   out_dir=str(tmp_path))
    assert tr2.resume()
    torch.manual_seed(0)
    out2 = tr2.gen.output(torch.rand(4, 2) * 0 + 0.5)
    out1 = tr.gen.output(torch.rand(4, 2) * 0 + 0.5)
    assert torch.allclose(out1, out2, atol=1e-6)
    # training continues from the restored state
    res = tr2.train_iteration(DataSet(feats, labels))
    assert torch.isfinite(torch.tensor(res["loss_d"]))


def test_resume_returns_false_without_checkpoints(tmp_path):
    cfg = GanConfig()
    cfg.train.use_gpu = False
    tr = ReferenceProtocolTrainer(cfg, device=torch.device("cpu"),
                                  out_dir=str(tmp_path / "empty"))
    assert not tr.resume()


def test_fast_trainer_save_resume(tmp_path):
    # GanTrainer checkpoint: params, buffers, updater state, EMA and
    # step counter all round-trip; training continues bit-consistently
    import torch
    from gan_deeplearning4j_amd.config import preset
    from gan_deeplearning4j_amd.models import build_mlp_gan
    from gan_deeplearning4j_amd.train import GanTrainer

    cfg = preset("mlp_tabular_cpu")
    cfg.train.use_gpu = False
    cfg.train.ema_decay = 0.9
    gen, dis = build_mlp_gan(cfg, hidden=16)
    tr = GanTrainer(gen, dis, cfg, device=torch.device("cpu"))
    x = torch.rand(8, cfg.data.num_features)
    for _ in range(2):
        tr.step(x)
    tr.save(tmp_path)

    gen2, dis2 = build_mlp_gan(cfg, hidden=16)
    tr2 = GanTrainer(gen2, dis2, cfg, device=torch.device("cpu"))
    assert tr2.resume(tmp_path)
    assert tr2.it == tr.it == 2
    assert torch.allclose(tr2.gen.params_flat(), tr.gen.params_flat())
    assert torch.allclose(tr2.dis.params_flat(), tr.dis.params_flat())
    for a, b in zip(tr._ema, tr2._ema):
        assert torch.allclose(a, b)
    assert tr2.gen.updater.t == tr.gen.updater.t
    # missing dir -> False, nothing touched
    gen3, dis3 = build_mlp_gan(cfg, hidden=16)
    tr3 = GanTrainer(gen3, dis3, cfg, device=torch.device("cpu"))
    assert not tr3.resume(tmp_path / "nope")
