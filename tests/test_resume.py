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
