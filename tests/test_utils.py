"""Utility-layer tests: imaging grid writer, seeding, metrics logger,
runtime config (CPU-only paths)."""

import json

import torch


def test_save_image_grid_rgb_and_gray(tmp_path):
    from gan_deeplearning4j_amd.utils.imaging import save_image_grid

    rgb = torch.rand(7, 3, 8, 8) * 2 - 1  # [-1,1] branch, ragged grid
    p = save_image_grid(rgb, tmp_path / "g.png", nrow=4)
    assert p.exists() and p.stat().st_size > 0
    gray = torch.rand(4, 1, 6, 6)  # [0,1] branch, single channel
    p2 = save_image_grid(gray, tmp_path / "sub" / "g2.png", nrow=2)
    assert p2.exists()  # parent dir auto-created


def test_seed_everything_reproducible():
    from gan_deeplearning4j_amd.utils.seed import seed_everything

    seed_everything(666)
    a = torch.randn(4)
    seed_everything(666)
    b = torch.randn(4)
    assert torch.equal(a, b)
    # ranks decorrelate (one stream per data-parallel worker)
    seed_everything(666, rank=1)
    c = torch.randn(4)
    assert not torch.equal(a, c)


def test_metrics_logger_jsonl(tmp_path):
    from gan_deeplearning4j_amd.train.metrics import MetricsLogger

    m = MetricsLogger(out_dir=str(tmp_path), print_every=10)
    m.step(0, loss_d=0.5, loss_g=1.25)
    m.step(1, loss_d=0.4, loss_g=1.0)
    m.close()
    lines = [json.loads(line) for line in
             (tmp_path / "metrics.jsonl").read_text().splitlines()]
    assert [r["iter"] for r in lines] == [0, 1]
    assert lines[0]["loss_g"] == 1.25
    assert all("step_time_s" in r for r in lines)


def test_metrics_logger_non_main_writes_nothing(tmp_path):
    from gan_deeplearning4j_amd.train.metrics import MetricsLogger

    m = MetricsLogger(out_dir=str(tmp_path), is_main=False)
    m.step(0, loss=1.0)
    m.close()
    assert not (tmp_path / "metrics.jsonl").exists()


def test_configure_runtime_cpu_noop():
    from gan_deeplearning4j_amd.utils.runtime import configure_runtime

    # must not raise without a GPU (reference's CudaEnvironment block is
    # GPU-only too, Java:103-115)
    configure_runtime(deterministic=True, verbose=False)


def test_memory_summary_cpu():
    from gan_deeplearning4j_amd.utils.runtime import memory_summary

    s = memory_summary()
    assert isinstance(s, dict)
