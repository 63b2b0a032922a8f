import sys

import numpy as np

from gan_deeplearning4j_amd.main import main


def test_reference_protocol_cli(tmp_path):
    out = main([
        "--protocol", "reference",
        f"data.data_dir={tmp_path}/data",
        f"train.out_dir={tmp_path}/out",
        "data.synthetic_size=64",
        "data.batch_size_per_worker=16",
        "data.batch_size_pred=32",
        "train.num_iterations=1",
        "train.use_gpu=false",
    ])
    assert np.isfinite(out["loss_d"])
    assert (tmp_path / "out" / "mnist_dis_model.zip").exists()
    assert (tmp_path / "out" / "mnist_out_1.csv").exists()
    assert (tmp_path / "out" / "generated_grid_1.png").exists()


def test_fast_protocol_cli(tmp_path):
    out = main([
        "--protocol", "fast",
        f"data.data_dir={tmp_path}/data",
        f"train.out_dir={tmp_path}/out",
        "model.arch=mlp",
        "model.z_size=16",
        "data.num_features=64",
        "data.label_index=64",
        "data.synthetic_kind=transactions",
        "data.synthetic_size=64",
        "data.batch_size_per_worker=16",
        "train.num_iterations=2",
        "train.use_gpu=false",
    ])
    assert out["iterations"] == 2
    assert (tmp_path / "out" / "sample_grid.csv").exists()
    assert (tmp_path / "out" / "gen_model.zip").exists()
    assert (tmp_path / "out" / "metrics.jsonl").exists()
