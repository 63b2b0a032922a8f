"""CLI entry point (the reference's `main` / `GAN()`, Java:94-101).

    python -m gan_deeplearning4j_amd.main [--config cfg.yaml] [--protocol
        reference|fast] [section.key=value ...]

- protocol=reference: the exact DL4J alternating protocol on CSV data
  (3 graphs + transfer classifier + CSV artifact dumps + 4 zip
  checkpoints per iteration).
- protocol=fast: the idiomatic MI355X trainer (DCGAN/MLP per
  model.arch) on synthetic or CSV data, DDP-ready (torchrun).
"""

from __future__ import annotations

import argparse
import logging
import sys
from pathlib import Path

import torch

from .config import GanConfig
from .data import CSVRecordReader, RecordReaderDataSetIterator, write_synthetic_csv
from .models import build_dcgan, build_mlp_gan
from .parallel.launch import init_distributed, is_main
from .train import GanTrainer, MetricsLogger, ReferenceProtocolTrainer

log = logging.getLogger("gan_deeplearning4j_amd")


def _iterators(cfg: GanConfig, rank: int = 0):
    """Train/test DataSet iterators. rank decorrelates the training
    shuffle order across data-parallel workers (every rank still sees
    the full dataset and steps in lockstep — no tail-imbalance deadlock
    on the gradient collectives; the reference's TrainingMaster
    partitions exported batches per worker, Java:325-330)."""
    d = cfg.data
    data_dir = Path(d.data_dir)
    train_csv = data_dir / f"{d.dataset_name}_train.csv"
    test_csv = data_dir / f"{d.dataset_name}_test.csv"
    if not train_csv.exists():
        if not d.synthetic:
            raise FileNotFoundError(train_csv)
        log.info("generating synthetic %s data -> %s", d.synthetic_kind, data_dir)
        kw = {}
        if d.synthetic_kind == "pixel_lattice":
            kw = dict(height=cfg.model.image_height, width=cfg.model.image_width,
                      channels=cfg.model.image_channels,
                      num_classes=d.num_classes)
        else:
            kw = dict(num_features=d.num_features, num_classes=d.num_classes)
        write_synthetic_csv(train_csv, d.synthetic_kind, n=d.synthetic_size,
                            seed=cfg.train.seed, **kw)
        write_synthetic_csv(test_csv, d.synthetic_kind,
                            n=max(100, d.synthetic_size // 5),
                            seed=cfg.train.seed + 1, **kw)
    train_it = RecordReaderDataSetIterator(
        CSVRecordReader().initialize(train_csv), d.batch_size_per_worker,
        d.label_index, d.num_classes, shuffle=True,
        seed=cfg.train.seed + 1000 * rank)
    test_it = RecordReaderDataSetIterator(
        CSVRecordReader().initialize(test_csv), d.batch_size_pred,
        d.label_index, d.num_classes)
    return train_it, test_it


def run_reference(cfg: GanConfig):
    train_it, test_it = _iterators(cfg)
    tr = ReferenceProtocolTrainer(cfg)
    out = tr.run(train_it, test_it)
    log.info("done: %s", out)
    return out


def run_fast(cfg: GanConfig):
    rank, world, local_rank, device = init_distributed(cfg.parallel.backend)
    if cfg.model.arch == "mlp":
        gen, dis = build_mlp_gan(cfg)
    else:
        gen, dis = build_dcgan(cfg)
    tr = GanTrainer(gen, dis, cfg, device=device)
    train_it, _ = _iterators(cfg, rank)
    metrics = MetricsLogger(cfg.train.out_dir, cfg.train.print_every,
                            is_main())
    # crash recovery: pick up from the last full checkpoint if present
    if tr.resume(cfg.train.out_dir):
        log.info("resumed fast trainer at iteration %d", tr.it)
    m = cfg.model
    it = tr.it
    while it < cfg.train.num_iterations:
        for ds in train_it:
            if it >= cfg.train.num_iterations:
                break
            feats = ds.features
            if m.arch != "mlp":
                feats = feats.reshape(-1, m.image_channels, m.image_height,
                                      m.image_width) * 2 - 1
            out = tr.step(feats)
            it += 1
            metrics.step(it, loss_d=float(out["loss_d"]),
                         loss_g=float(out["loss_g"]), images=out["images"])
            if is_main() and it % cfg.train.save_every == 0:
                tr.save(cfg.train.out_dir)
    if is_main():
        import numpy as np

        grid = tr.sample_grid(cfg.train.num_gen_samples)
        np.savetxt(Path(cfg.train.out_dir) / "sample_grid.csv",
                   grid.reshape(grid.shape[0], -1).numpy(), delimiter=",",
                   fmt="%.6f")
    return {"iterations": it}


def main(argv=None):
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(message)s")
    ap = argparse.ArgumentParser(description="gan_deeplearning4j_amd trainer")
    ap.add_argument("--config", type=str, default=None)
    ap.add_argument("--protocol", choices=("reference", "fast"),
                    default="reference")
    ap.add_argument("overrides", nargs="*")
    ns = ap.parse_args(argv)
    cfg = GanConfig.from_yaml(ns.config) if ns.config else GanConfig()
    cfg = cfg.apply_overrides(ns.overrides)
    if ns.protocol == "reference":
        return run_reference(cfg)
    return run_fast(cfg)


if __name__ == "__main__":
    main(sys.argv[1:])
