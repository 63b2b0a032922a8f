"""Configuration system.

The reference hard-codes 27 knobs as `private static final` constants
(reference Java dl4jGANComputerVision.java:66-92).  Here they are a real
config: dataclasses with YAML load/save and CLI overrides.
"""

from __future__ import annotations

import argparse
import dataclasses
import json
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, Optional

try:
    import yaml  # type: ignore

    _HAVE_YAML = True
except Exception:  # pragma: no cover
    _HAVE_YAML = False


@dataclass
class DataConfig:
    """Data pipeline knobs (reference Java:66-67, 85-90; notebook cell 2)."""

    dataset_name: str = "mnist"            # Java:90 dataSetName
    data_dir: str = "data"                 # Java:85-88 path constants
    batch_size_per_worker: int = 200       # Java:66 batchSizePerWorker
    batch_size_pred: int = 500             # Java:67 batchSizePred
    label_index: int = 784                 # Java:68 labelIndex
    num_classes: int = 10                  # Java:69 numClasses
    num_features: int = 784                # Java:71 numFeatures
    # Synthetic generators (no-network environment):
    synthetic: bool = True
    synthetic_kind: str = "pixel_lattice"  # or "transactions"
    synthetic_size: int = 10000


@dataclass
class ModelConfig:
    """Model topology knobs (reference Java:78-84)."""

    arch: str = "dcgan28"          # dcgan28 | dcgan64 | dcgan128 | mlp
    image_height: int = 28         # Java:80 height
    image_width: int = 28          # Java:81 width
    image_channels: int = 1        # Java:82 channels
    z_size: int = 2                # Java:84 zSize
    num_classes_dis: int = 1       # Java:70 numClassesDis (binary real/fake)
    base_width: int = 64           # conv channel multiplier (dcgan64/128)
    dtype: str = "bf16"            # compute dtype on GPU


@dataclass
class OptimConfig:
    """Updater knobs (reference Java:74-76, 123-127, 133-160, 233-243)."""

    optimizer: str = "adam"                 # "adam" (north-star) or "rmsprop" (reference)
    dis_learning_rate: float = 2e-3         # Java:74 dis_learning_rate
    gen_learning_rate: float = 4e-3         # Java:75 gen_learning_rate
    frozen_learning_rate: float = 0.0       # Java:76 frozen_learning_rate
    beta1: float = 0.5
    beta2: float = 0.999
    rms_decay: float = 0.95
    epsilon: float = 1e-8                   # Java RMSProp epsilon
    grad_clip: float = 1.0                  # ClipElementWiseAbsoluteValue(1.0), Java:123-124
    l2: float = 1e-4                        # Java:125 l2(0.0001)
    weight_init: str = "xavier"             # Java:127 WeightInit.XAVIER
    lr_schedule: str = ""                   # framework extension: "" (fixed,
    #                                         reference semantics) | "linear"
    #                                         | "cosine" over num_iterations
    lr_warmup_steps: int = 0                # linear warmup before the decay


@dataclass
class TrainConfig:
    """Training-loop knobs (reference Java:72-77, 325-330, 405-406)."""

    num_iterations: int = 2            # Java:72 numIterations
    num_gen_samples: int = 10          # Java:73 numGenSamples (10x10 grid)
    seed: int = 666                    # Java:75 numberOfTheBeast
    print_every: int = 1               # Java:77 printEvery
    save_every: int = 1                # Java:77 saveEvery
    label_noise_std: float = 0.05      # one-sided label softening, Java:405-406
    averaging_frequency: int = 10      # ParameterAveragingTrainingMaster, Java:326
    worker_prefetch_num_batches: int = 0  # Java:328
    out_dir: str = "out"
    use_gpu: bool = True               # Java:92 useGpu
    reference_semantics: bool = False  # exact frozen-copy + weight-sync protocol
    train_classifier: bool = True      # transfer-learned classifier path
    ema_decay: float = 0.0             # >0: keep an fp32 EMA of G's params
    #                                    (framework extension; 0 = off,
    #                                    matching the reference's protocol)
    loss_type: str = "bce"             # adversarial objective of the fast
    #                                    trainer: bce (reference XENT) |
    #                                    lsgan (least-squares) | hinge
    augment: str = ""                  # DiffAugment policy for D inputs,
    #                                    e.g. "translate,cutout" ("" = off)
    d_steps_per_g: int = 1             # n-critic: D updates per G update


@dataclass
class ParallelConfig:
    """Distributed knobs (MI355X-native: one process/GPU over RCCL/xGMI)."""

    backend: str = "auto"        # auto -> nccl(=RCCL) on GPU, gloo on CPU
    bucket_cap_mb: int = 40      # gradient allreduce bucket size (xGMI-tuned)
    overlap_grad_reduce: bool = True
    local_steps: int = 1         # averaging_frequency-style local steps


@dataclass
class GanConfig:
    data: DataConfig = field(default_factory=DataConfig)
    model: ModelConfig = field(default_factory=ModelConfig)
    optim: OptimConfig = field(default_factory=OptimConfig)
    train: TrainConfig = field(default_factory=TrainConfig)
    parallel: ParallelConfig = field(default_factory=ParallelConfig)

    # ------------------------------------------------------------------ io
    def to_dict(self) -> dict:
        return dataclasses.asdict(self)

    def to_yaml(self, path: str | Path) -> None:
        text = (
            yaml.safe_dump(self.to_dict(), sort_keys=False)
            if _HAVE_YAML
            else json.dumps(self.to_dict(), indent=2)
        )
        Path(path).write_text(text)

    @classmethod
    def from_dict(cls, d: dict) -> "GanConfig":
        def mk(klass, sub: Optional[dict]):
            sub = sub or {}
            names = {f.name for f in dataclasses.fields(klass)}
            return klass(**{k: v for k, v in sub.items() if k in names})

        return cls(
            data=mk(DataConfig, d.get("data")),
            model=mk(ModelConfig, d.get("model")),
            optim=mk(OptimConfig, d.get("optim")),
            train=mk(TrainConfig, d.get("train")),
            parallel=mk(ParallelConfig, d.get("parallel")),
        )

    @classmethod
    def from_yaml(cls, path: str | Path) -> "GanConfig":
        text = Path(path).read_text()
        d = yaml.safe_load(text) if _HAVE_YAML else json.loads(text)
        return cls.from_dict(d)

    # ----------------------------------------------------------------- cli
    def apply_overrides(self, overrides: list[str]) -> "GanConfig":
        """Apply 'section.key=value' overrides (CLI)."""
        d = self.to_dict()
        for ov in overrides:
            key, _, raw = ov.partition("=")
            if not _:
                raise ValueError(f"override must be section.key=value: {ov!r}")
            sect, _, name = key.partition(".")
            if sect not in d or name not in d[sect]:
                raise KeyError(f"unknown config key: {key}")
            cur = d[sect][name]
            d[sect][name] = _coerce(raw, type(cur))
        return GanConfig.from_dict(d)

    @classmethod
    def from_cli(cls, argv: Optional[list[str]] = None) -> "GanConfig":
        ap = argparse.ArgumentParser(description="gan_deeplearning4j_amd")
        ap.add_argument("--config", type=str, default=None, help="YAML config file")
        ap.add_argument("overrides", nargs="*", help="section.key=value overrides")
        ns = ap.parse_args(argv)
        cfg = cls.from_yaml(ns.config) if ns.config else cls()
        return cfg.apply_overrides(ns.overrides)


def _coerce(raw: str, t: type) -> Any:
    if t is bool:
        return raw.lower() in ("1", "true", "yes", "on")
    if t is int:
        return int(raw)
    if t is float:
        return float(raw)
    return raw


# Preset configs matching BASELINE.json's five configs.
def preset(name: str) -> GanConfig:
    cfg = GanConfig()
    if name == "mlp_tabular_cpu":
        cfg.model = ModelConfig(arch="mlp", image_height=1, image_width=1,
                                image_channels=1, z_size=16)
        cfg.data.synthetic_kind = "transactions"
        cfg.data.num_features = 64
        cfg.data.label_index = 64
        cfg.train.use_gpu = False
    elif name == "dcgan28":
        cfg.model = ModelConfig(arch="dcgan28", image_height=28, image_width=28,
                                image_channels=1, z_size=2)
    elif name == "dcgan64":
        cfg.model = ModelConfig(arch="dcgan64", image_height=64, image_width=64,
                                image_channels=3, z_size=128, base_width=64)
        cfg.data.num_features = 64 * 64 * 3
        cfg.data.label_index = 64 * 64 * 3
    elif name == "dcgan128":
        cfg.model = ModelConfig(arch="dcgan128", image_height=128, image_width=128,
                                image_channels=3, z_size=128, base_width=64,
                                dtype="fp8")
        cfg.data.num_features = 128 * 128 * 3
        cfg.data.label_index = 128 * 128 * 3
    else:
        raise KeyError(f"unknown preset {name!r}")
    return cfg
