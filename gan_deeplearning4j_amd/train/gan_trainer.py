"""GAN trainers.

Two protocols over the same op library:

- `GanTrainer` — the idiomatic MI355X fast path used by the flagship
  bench: shared G/D graphs, proper stop-gradient freezing during the
  G-step (no frozen twin graphs, no weight-copy passes), fused updaters,
  bucketed RCCL gradient all-reduce overlapped with backward.

- `ReferenceProtocolTrainer` — the reference's exact alternating protocol
  (Java:408-621): three graphs (D, frozen gen, stacked gan), one-shot
  label softening, manual per-tensor weight sync between twins,
  transfer-learned classifier trained alongside, CSV artifact dumps and
  4x DL4J-zip checkpoints per iteration, optional parameter-averaging
  distributed semantics (averaging_frequency).
"""

from __future__ import annotations

import logging
from contextlib import contextmanager
from pathlib import Path
from typing import Optional

import numpy as np
import torch

from ..config import GanConfig
from ..data.csv_reader import DataSet
from ..graph.serialization import ModelSerializer
from ..models import (
    DIS_TO_CV_SYNC,
    DIS_TO_GAN_SYNC,
    GAN_TO_GEN_SYNC,
    build_discriminator,
    build_frozen_generator,
    build_stacked_gan,
    build_transfer_classifier,
)
from ..models.reference_protocol import sync_params
from ..ops import functional as OF
from ..parallel.ddp import GradReducer, average_parameters, broadcast_parameters
from .metrics import MetricsLogger

log = logging.getLogger("gan_deeplearning4j_amd")


def latent_grid(n: int, z_size: int, device="cpu", dtype=torch.float32):
    """n x n grid over linspace(-1,1)^2 (reference Java:382-389).
    For z_size > 2 the remaining dims are zero."""
    lin = torch.linspace(-1, 1, n)
    zs = torch.zeros(n * n, z_size)
    for i in range(n):
        for j in range(n):
            zs[i * n + j, 0] = lin[i]
            zs[i * n + j, min(1, z_size - 1)] = lin[j]
    return zs.to(device=device, dtype=dtype)


class GanTrainer:
    """Idiomatic alternating GAN trainer (flagship/bench path)."""

    def __init__(
        self,
        gen,
        dis,
        cfg: GanConfig,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
        bucket_cap_mb: Optional[int] = None,
        capture: bool = False,
    ):
        self.cfg = cfg
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() and cfg.train.use_gpu else "cpu"
        )
        if dtype is None:
            dtype = (
                torch.bfloat16
                if self.device.type == "cuda" and cfg.model.dtype in ("bf16", "fp8")
                else torch.float32
            )
        self.dtype = dtype
        self.gen = gen.to_device(self.device, dtype)
        self.dis = dis.to_device(self.device, dtype)
        broadcast_parameters(self.gen)
        broadcast_parameters(self.dis)
        cap = bucket_cap_mb or cfg.parallel.bucket_cap_mb
        # the G-step reducer is the LAST to finish each step, so it also
        # averages BOTH models' BN running stats (D's update again during
        # the G-step forward, after d_reducer.finish)
        self.g_reducer = GradReducer([self.gen], cap, cfg.parallel.local_steps,
                                     buffer_modules=[self.gen, self.dis])
        self.d_reducer = GradReducer([self.dis], cap, cfg.parallel.local_steps,
                                     sync_buffers=False)
        self.z_size = cfg.model.z_size
        from ..parallel.launch import get_rank

        self._g = torch.Generator(device="cpu").manual_seed(
            cfg.train.seed + 1000 * get_rank()
        )
        # one-sided label softening, drawn once (reference Java:405-406)
        std = cfg.train.label_noise_std
        self._soft_real = None
        self._soft_fake = None
        self._noise_std = std
        self.it = 0
        # D-step batching: separate real/fake forwards (default; per-half
        # BatchNorm statistics, the conventional DCGAN recipe and the
        # reference's semantics) vs one concatenated 2n forward
        # (GDLJ_D_CONCAT=1: half the D launches at twice the GEMM M,
        # mixed-batch BN statistics — a legitimate alternative recipe,
        # kept opt-in pending a quality + throughput A/B).
        import os as _os

        self._d_concat = _os.environ.get("GDLJ_D_CONCAT") == "1"
        # adversarial objective (framework extension; the reference is
        # sigmoid+XENT = bce): lsgan = least-squares on logits, hinge =
        # margin loss (label softening applies to bce/lsgan targets only)
        self.loss_type = getattr(cfg.train, "loss_type", "bce")
        if self.loss_type not in ("bce", "lsgan", "hinge"):
            raise ValueError(f"unknown loss_type {self.loss_type!r}")
        # DiffAugment policy for D inputs (train/augment.py; "" = off)
        self.augment = getattr(cfg.train, "augment", "")
        if self.augment:
            from .augment import diff_augment as _da

            _da(torch.zeros(1, 1, 8, 8), self.augment)  # validate policy
        if self.loss_type != "bce" or self.augment:
            self._d_concat = False  # concat path is plain-bce only
        # fp32 EMA of G's params (framework extension; sampling/serving
        # quality knob standard in production GAN trainers). Updated
        # after every G step — inside the captured graph when capturing.
        self.ema_decay = float(getattr(cfg.train, "ema_decay", 0.0))
        self._ema = ([p.detach().float().clone()
                      for p in self.gen.parameters()]
                     if self.ema_decay > 0 else None)
        # n-critic: D updates per G update (1 = the reference recipe)
        self.d_steps = max(1, int(getattr(cfg.train, "d_steps_per_g", 1)))
        # LR schedule (framework extension): warmup + linear/cosine decay
        # over num_iterations. Incompatible with graph capture (lr is a
        # scalar kernel argument, frozen at capture) -> forces eager.
        self.lr_schedule = getattr(cfg.optim, "lr_schedule", "")
        self.lr_warmup = int(getattr(cfg.optim, "lr_warmup_steps", 0))
        if self.lr_schedule not in ("", "linear", "cosine"):
            raise ValueError(f"unknown lr_schedule {self.lr_schedule!r}")
        self._horizon = max(int(cfg.train.num_iterations), 1)
        # hipGraph capture of the whole training step (one replay per step;
        # removes launch/dispatch host overhead). GPU-only; falls back to
        # eager if capture fails.
        self.capture = capture and self.device.type == "cuda" \
            and not self.lr_schedule and self.lr_warmup == 0
        self._graph = None
        self._graph_failed = False
        self._static_real = None
        self._graph_out = None
        if self.device.type == "cuda":
            torch.cuda.manual_seed(cfg.train.seed + 1000 * get_rank())
            if cfg.model.dtype == "fp8":
                from ..ops.gpu_ops import set_fp8_conv

                set_fp8_conv(True)

    def _labels(self, n: int):
        if self._soft_real is None or self._soft_real.shape[0] != n:
            std = self._noise_std
            self._soft_real = (
                1.0 + std * torch.randn(n, 1, generator=self._g)
            ).to(self.device)
            self._soft_fake = (std * torch.randn(n, 1, generator=self._g)).to(
                self.device
            )
        return self._soft_real, self._soft_fake

    def sample_z(self, n: int) -> torch.Tensor:
        if self.device.type == "cuda":
            # device RNG: capture-aware (philox offset advances per replay)
            return torch.randn(n, self.z_size, device=self.device,
                               dtype=self.dtype)
        z = torch.randn(n, self.z_size, generator=self._g)
        return z.to(self.device, self.dtype)

    def _d_loss(self, real, fake, soft_real, soft_fake, cond=()):
        if cond:
            d_real = self.dis(real, *cond)
            d_fake = self.dis(fake.detach(), *cond)
            if self.loss_type == "lsgan":
                return OF.mse_loss(d_real, soft_real) + \
                    OF.mse_loss(d_fake, soft_fake)
            if self.loss_type == "hinge":
                return torch.relu(1.0 - d_real.float()).mean() + \
                    torch.relu(1.0 + d_fake.float()).mean()
            return OF.bce_with_logits_loss(d_real, soft_real) + \
                OF.bce_with_logits_loss(d_fake, soft_fake)
        if self._d_concat:
            both = torch.cat([real, fake.detach()], dim=0)
            d_all = self.dis(both)
            lab = torch.cat([soft_real, soft_fake], dim=0)
            # x2 keeps the gradient scale of mean(real)+mean(fake)
            return OF.bce_with_logits_loss(d_all, lab) * 2
        real_in, fake_in = real, fake.detach()
        if self.augment and real.dim() == 4:  # images only
            from .augment import diff_augment

            real_in = diff_augment(real_in, self.augment)
            fake_in = diff_augment(fake_in, self.augment)
        d_real = self.dis(real_in)
        d_fake = self.dis(fake_in)
        if self.loss_type == "lsgan":
            return OF.mse_loss(d_real, soft_real) + \
                OF.mse_loss(d_fake, soft_fake)
        if self.loss_type == "hinge":
            return torch.relu(1.0 - d_real.float()).mean() + \
                torch.relu(1.0 + d_fake.float()).mean()
        return OF.bce_with_logits_loss(d_real, soft_real) + \
            OF.bce_with_logits_loss(d_fake, soft_fake)

    def lr_scale_at(self, it: int) -> float:
        """Schedule multiplier for 1-based step `it` (1.0 when off)."""
        if not self.lr_schedule and self.lr_warmup == 0:
            return 1.0
        if self.lr_warmup > 0 and it <= self.lr_warmup:
            return it / self.lr_warmup
        if not self.lr_schedule:
            return 1.0
        span = max(self._horizon - self.lr_warmup, 1)
        prog = min(max(it - self.lr_warmup, 0) / span, 1.0)
        if self.lr_schedule == "linear":
            return 1.0 - prog
        import math

        return 0.5 * (1.0 + math.cos(math.pi * prog))  # cosine

    def step(self, real: torch.Tensor,
             labels: Optional[torch.Tensor] = None) -> dict:
        """One alternating D+G update on a batch of real images.

        labels: conditioning input for multi-input (cGAN) graphs —
        passed as the second graph input to both G and D. Conditional
        steps always run eager (capture is keyed to one input shape).
        """
        self.it += 1
        if self.lr_schedule or self.lr_warmup:
            scale = self.lr_scale_at(self.it)
            self.gen.updater.lr_scale = scale
            self.dis.updater.lr_scale = scale
        if labels is None and self.capture and not self._graph_failed:
            return self._step_graphed(real)
        return self._step_eager(real, labels)

    def _step_eager(self, real: torch.Tensor,
                    labels: Optional[torch.Tensor] = None) -> dict:
        n = real.shape[0]
        real = real.to(self.device, self.dtype)
        cond = () if labels is None else (
            labels.to(self.device, self.dtype),)
        soft_real, soft_fake = self._labels(n)

        self.gen.train()
        self.dis.train()

        # ---- D step(s): train.d_steps_per_g > 1 is the n-critic
        # recipe (fresh z per D update; the G step reuses the last
        # fake). d_steps == 1 reproduces the reference flow exactly. --
        fake = loss_d = None
        for _ in range(self.d_steps):
            z = self.sample_z(n)
            fake = self.gen(z, *cond)
            self.dis.updater.zero_grad()
            self.d_reducer.prepare()
            loss_d = self._d_loss(real, fake, soft_real, soft_fake, cond)
            loss_d.backward()
            self.d_reducer.finish()
            self.dis.updater.step()
        # ---- G step (stop-gradient freeze of D: no D wgrad compute) --
        for p in self.dis.parameters():
            p.requires_grad_(False)
        self.gen.updater.zero_grad()
        self.g_reducer.prepare()
        g_in = fake
        if self.augment and fake.dim() == 4 and not cond:
            from .augment import diff_augment

            g_in = diff_augment(g_in, self.augment)
        g_logits = self.dis(g_in, *cond)
        if self.loss_type == "lsgan":
            loss_g = OF.mse_loss(g_logits,
                                 torch.ones(n, 1, device=self.device))
        elif self.loss_type == "hinge":
            loss_g = -g_logits.float().mean()
        else:
            loss_g = OF.bce_with_logits_loss(
                g_logits, torch.ones(n, 1, device=self.device)
            )
        loss_g.backward()
        self.g_reducer.finish()
        self.gen.updater.step()
        for p in self.dis.parameters():
            p.requires_grad_(True)
        if self._ema is not None:
            with torch.no_grad():
                d = self.ema_decay
                for e, p in zip(self._ema, self.gen.parameters()):
                    e.mul_(d).add_(p.detach().to(e.dtype), alpha=1.0 - d)

        # losses stay on-device (no .item() sync in the hot loop); callers
        # float() them when they actually need host values
        return {
            "loss_d": loss_d.detach(),
            "loss_g": loss_g.detach(),
            "images": n,
        }

    # ----------------------------------------------------- hipGraph path
    def _invalidate_packed(self):
        for g in (self.gen, self.dis):
            for p in g.parameters():
                if hasattr(p, "_gdlj_cache"):
                    del p._gdlj_cache

    def _step_graphed(self, real: torch.Tensor) -> dict:
        if self._graph is None:
            try:
                self._capture(real)
            except Exception as e:  # fall back to eager permanently
                log.warning("hipGraph capture failed (%s); eager fallback", e)
                self._graph_failed = True
                self._graph = None
                return self._step_eager(real)
        if real.shape != self._static_real.shape:
            # shape changed after capture: run this batch eagerly
            return self._step_eager(real)
        self._static_real.copy_(real.to(self.device, self.dtype),
                                non_blocking=True)
        self._graph.replay()
        out = dict(self._graph_out)
        out["images"] = real.shape[0]
        return out

    def _capture(self, real: torch.Tensor):
        # warm up state (updater slots, labels, allocator) on a side stream
        self._static_real = real.to(self.device, self.dtype).clone()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._step_eager(self._static_real)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        # caches must be rebuilt INSIDE the capture so the re-pack kernels
        # (reading freshly-updated params) are part of the recorded stream
        self._invalidate_packed()
        self._graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self._graph):
            self._graph_out = self._step_eager(self._static_real)
        log.info("training step captured as hipGraph")

    # ------------------------------------------------- save / resume
    def save(self, out_dir) -> None:
        """Checkpoint both graphs (DL4J zip layout + updater state),
        the EMA state and the step counter. The reference protocol
        checkpoints per-iteration (Java:605-618); this is the fast
        trainer's equivalent."""
        out = Path(out_dir)
        out.mkdir(parents=True, exist_ok=True)
        ModelSerializer.write_model(self.gen, out / "gen_model.zip",
                                    save_updater=True)
        ModelSerializer.write_model(self.dis, out / "dis_model.zip",
                                    save_updater=True)
        state = {"it": self.it}
        if self._ema is not None:
            state["ema"] = [e.cpu() for e in self._ema]
        torch.save(state, out / "trainer_state.pt")

    def resume(self, out_dir) -> bool:
        """Restore params, buffers, updater state, EMA and the step
        counter IN PLACE (tensor identities survive, so captured
        graphs, reducers and packed-weight caches stay valid — the
        caches re-key on the bumped tensor versions)."""
        out = Path(out_dir)
        paths = [out / "gen_model.zip", out / "dis_model.zip",
                 out / "trainer_state.pt"]
        if not all(p.exists() for p in paths):
            return False
        for live, path in ((self.gen, paths[0]), (self.dis, paths[1])):
            g2 = ModelSerializer.restore_computation_graph(
                path, load_updater=True)
            g2.to_device(self.device, self.dtype)
            with torch.no_grad():
                for p, p2 in zip(live.parameters(), g2.parameters()):
                    p.copy_(p2)
                for b, b2 in zip(live.buffers(), g2.buffers()):
                    b.copy_(b2)
            live.updater.load_state_dict(g2.updater.state_dict())
        state = torch.load(out / "trainer_state.pt", weights_only=True)
        self.it = int(state["it"])
        if self._ema is not None and "ema" in state:
            with torch.no_grad():
                for e, s in zip(self._ema, state["ema"]):
                    e.copy_(s.to(e.device))
        return True

    @contextmanager
    def ema_weights(self):
        """Temporarily swap the fp32 EMA weights into G (for sampling /
        serving / checkpointing the smoothed generator). No-op when
        ema_decay == 0. The packed-kernel caches key on tensor versions,
        so they rebuild for the swapped weights and again on restore."""
        if self._ema is None:
            yield self.gen
            return
        backup = [p.detach().clone() for p in self.gen.parameters()]
        with torch.no_grad():
            for p, e in zip(self.gen.parameters(), self._ema):
                p.copy_(e.to(p.dtype))
        try:
            yield self.gen
        finally:
            with torch.no_grad():
                for p, b in zip(self.gen.parameters(), backup):
                    p.copy_(b)

    @torch.no_grad()
    def sample_grid(self, n: int = 10, ema: bool = False) -> torch.Tensor:
        z = latent_grid(n, self.z_size, self.device, self.dtype)
        if ema and self._ema is not None:
            with self.ema_weights():
                return self.gen.output(z).float().cpu()
        return self.gen.output(z).float().cpu()


class ReferenceProtocolTrainer:
    """The reference's exact 6-phase alternating loop (Java:408-621)."""

    def __init__(self, cfg: GanConfig, device: Optional[torch.device] = None,
                 out_dir: Optional[str] = None):
        self.cfg = cfg
        # Exact-reference updater numerics: the reference constructs
        # RmsProp(lr, 1e-8, 1e-8) (Java:133) — rmsDecay=1e-8, an
        # effectively memoryless v.  The dataclass default (0.95) is the
        # idiomatic value for the fast trainer; in reference-protocol
        # mode with rmsprop selected and the default untouched, use the
        # reference's constant.
        if (cfg.optim.optimizer == "rmsprop"
                and cfg.optim.rms_decay == 0.95):
            cfg.optim.rms_decay = 1e-8
        self.device = device or torch.device(
            "cuda" if torch.cuda.is_available() and cfg.train.use_gpu else "cpu"
        )
        self.out_dir = Path(out_dir or cfg.train.out_dir)
        self.out_dir.mkdir(parents=True, exist_ok=True)

        # graphs (Java:118-314, 337-368); bf16 compute on GPU, fp32 on CPU
        dt = torch.bfloat16 if self.device.type == "cuda" else None
        self.dis = build_discriminator(cfg).to_device(self.device, dt)
        self.gen = build_frozen_generator(cfg).to_device(self.device, dt)
        self.gan = build_stacked_gan(cfg).to_device(self.device, dt)
        self.cv = build_transfer_classifier(self.dis, cfg).to_device(
            self.device, dt)
        for g in (self.dis, self.gen, self.gan, self.cv):
            broadcast_parameters(g)

        # init-time summary + shape smoke forward, like the reference
        # prints after each graph init (Java:167-170, 223-225, 312-314,
        # 365-368): summary() then output(randn(10, n_in)).shape
        self.smoke_shapes = self._init_smoke()

        self.metrics = MetricsLogger(str(self.out_dir),
                                     cfg.train.print_every)
        torch.manual_seed(cfg.train.seed)
        self._cpu_gen = torch.Generator().manual_seed(cfg.train.seed)
        # label softening drawn ONCE before the loop (Java:405-406)
        b = cfg.data.batch_size_per_worker
        std = cfg.train.label_noise_std
        self.soft_fake = (std * torch.randn(b, 1, generator=self._cpu_gen))
        self.soft_real = (1 + std * torch.randn(b, 1, generator=self._cpu_gen))
        self.batch_counter = 0

    def _init_smoke(self) -> dict:
        """Reference parity: after init, print each graph's summary()
        and a 10-sample forward's output shape (Java:167-170, 223-225,
        312-314, 365-368). Returns {name: shape} for tests."""
        import logging

        log = logging.getLogger("gan_deeplearning4j_amd")
        nf = self.cfg.data.num_features
        zs = self.cfg.model.z_size
        shapes = {}
        specs = [("dis", self.dis, nf), ("gen", self.gen, zs),
                 ("gan", self.gan, zs), ("cv", self.cv, nf)]
        dt = (torch.bfloat16 if self.device.type == "cuda"
              else torch.float32)
        # dedicated generator: must not consume from the label-noise /
        # training RNG streams (golden trajectories depend on them)
        smoke_gen = torch.Generator().manual_seed(
            self.cfg.train.seed + 10)
        for name, g, n_in in specs:
            log.debug("%s summary:\n%s", name, g.summary())
            x = torch.randn(10, n_in, generator=smoke_gen).to(
                self.device, dt)
            g.eval()
            with torch.no_grad():
                y = g.output(x)
            g.train()
            shapes[name] = tuple(y.shape)
            log.info("%s smoke forward: (10, %d) -> %s", name, n_in,
                     shapes[name])
        return shapes

    # ----------------------------------------------------------- resume
    def resume(self) -> bool:
        """Restore all four graphs (incl. updater state) from the zip
        checkpoints in out_dir, if present. The reference is save-only
         — resume is this framework's extension."""
        paths = {name: self.out_dir / f"mnist_{name}_model.zip"
                 for name in ("dis", "gan", "gen", "CV")}
        if not all(p.exists() for p in paths.values()):
            return False
        dt = torch.bfloat16 if self.device.type == "cuda" else None
        self.dis = ModelSerializer.restore_computation_graph(
            paths["dis"]).to_device(self.device, dt)
        self.gan = ModelSerializer.restore_computation_graph(
            paths["gan"]).to_device(self.device, dt)
        self.gen = ModelSerializer.restore_computation_graph(
            paths["gen"]).to_device(self.device, dt)
        self.cv = ModelSerializer.restore_computation_graph(
            paths["CV"]).to_device(self.device, dt)
        log.info("resumed 4 graphs from %s", self.out_dir)
        return True

    # ------------------------------------------------------------------
    def _uniform_z(self, n: int) -> torch.Tensor:
        # Nd4j.rand.muli(2).subi(1): U(-1,1) (Java:420, 465)
        return (torch.rand(n, self.cfg.model.z_size,
                           generator=self._cpu_gen) * 2 - 1)

    def train_iteration(self, real: DataSet) -> dict:
        """One full reference iteration on a real labeled batch."""
        cfg = self.cfg
        dev = self.device
        n = real.num_examples()
        feats = real.features.to(dev)

        # (a) D-step on {real -> soft 1, G(z) -> soft 0} (Java:408-426)
        z = self._uniform_z(n).to(dev)
        fake_imgs = self.gen.output(z)
        fake_flat = fake_imgs.reshape(n, -1)
        d_data = [
            DataSet(feats, self.soft_real[:n].to(dev)),
            DataSet(fake_flat, self.soft_fake[:n].to(dev)),
        ]
        # the reference parallelizes the two DataSets across workers and
        # parameter-averages (Java:425-426): ONE averaged update, not two
        # sequential minibatch steps
        loss_d = self.dis.fit_averaged(d_data)

        # (b) copy fresh D into gan's frozen D (Java:429-460)
        sync_params(self.dis, self.gan, DIS_TO_GAN_SYNC)

        # (c) G-step through the stacked gan: {z~U(-1,1) -> 1} (Java:462-471)
        z2 = self._uniform_z(n).to(dev)
        loss_g = self.gan.fit(DataSet(z2, torch.ones(n, 1, device=dev)))

        # (d) copy trained G from gan into frozen gen (Java:474-510)
        sync_params(self.gan, self.gen, GAN_TO_GEN_SYNC)

        # (e) copy D backbone into classifier (Java:516-542)
        sync_params(self.dis, self.cv, DIS_TO_CV_SYNC)

        # (f) classifier step on the real labeled batch (Java:544-545)
        loss_cv = (
            self.cv.fit(DataSet(feats, real.labels.to(dev)))
            if cfg.train.train_classifier
            else 0.0
        )

        self.batch_counter += 1
        # parameter averaging every averaging_frequency batches (Java:326)
        if cfg.train.averaging_frequency > 0 and \
                self.batch_counter % cfg.train.averaging_frequency == 0:
            for g in (self.dis, self.gan, self.cv):
                average_parameters(g)
            sync_params(self.gan, self.gen, GAN_TO_GEN_SYNC)

        out = {"loss_d": loss_d, "loss_g": loss_g, "loss_cv": loss_cv,
               "images": n}
        self.metrics.step(self.batch_counter, **out)
        return out

    # ------------------------------------------------- artifacts (L5)
    @torch.no_grad()
    def dump_sample_grid(self, idx: int) -> Path:
        """10x10 latent-manifold grid -> mnist_out_<i>.csv (Java:550-570).

        The reference had a writer bug (fileWriter.close() inside the row
        loop, Java:568-569) — intent replicated, bug not."""
        n = self.cfg.train.num_gen_samples
        z = latent_grid(n, self.cfg.model.z_size, self.device)
        img4d = self.gen.output(z).float().cpu()
        imgs = img4d.reshape(n * n, -1)
        path = self.out_dir / f"mnist_out_{idx}.csv"
        np.savetxt(path, imgs.numpy(), delimiter=",", fmt="%.6f")
        # the notebook's tiled PNG (DCGAN_Generated_Images.png analog)
        try:
            from ..utils.imaging import save_image_grid

            save_image_grid(img4d, self.out_dir / f"generated_grid_{idx}.png",
                            nrow=n)
        except Exception:  # pragma: no cover - matplotlib optional
            pass
        return path

    @torch.no_grad()
    def dump_test_predictions(self, idx: int, test_iter) -> Path:
        """Classifier softmax on the test fold -> CSV (Java:572-598)."""
        rows = []
        for ds in test_iter:
            probs = self.cv.output(ds.features.to(self.device))
            rows.append(probs.float().cpu().numpy())
        path = self.out_dir / f"mnist_test_predictions_{idx}.csv"
        np.savetxt(path, np.concatenate(rows), delimiter=",", fmt="%.6f")
        return path

    def save_checkpoints(self, save_updater: bool = True) -> list[Path]:
        """4x DL4J-zip checkpoints (Java:605-618)."""
        ckpts = []
        for name, g in (("dis", self.dis), ("gan", self.gan),
                        ("gen", self.gen), ("CV", self.cv)):
            p = self.out_dir / f"mnist_{name}_model.zip"
            ModelSerializer.write_model(g, p, save_updater=save_updater)
            ckpts.append(p)
        return ckpts

    # ---------------------------------------------------------- driver
    def run(self, train_iter, test_iter=None) -> dict:
        """The while-loop (Java:408): numIterations batches (re-iterating
        the data epoch-wise until the budget is met)."""
        cfg = self.cfg
        last = {}
        while self.batch_counter < cfg.train.num_iterations:
            progressed = False
            for ds in train_iter:
                if self.batch_counter >= cfg.train.num_iterations:
                    break
                progressed = True
                last = self.train_iteration(ds)
                i = self.batch_counter
                if i % cfg.train.print_every == 0:
                    self.dump_sample_grid(i)
                    if test_iter is not None and cfg.train.train_classifier:
                        self.dump_test_predictions(i, test_iter)
                if i % cfg.train.save_every == 0:
                    self.save_checkpoints()
            if not progressed:
                break
        return last
