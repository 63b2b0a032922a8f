"""GraphBuilder / ComputationGraph — the DL4J-style model API.

Recreates the API surface the reference exercises:
NeuralNetConfiguration.Builder().graphBuilder() with addInputs /
setInputTypes / addLayer / inputPreProcessor / setOutputs / build, then
init(), summary(), output(), fit(), getLayer(name).getParam/setParam
(reference Java:118-165, 173-221, 228-310).

Internally a ComputationGraph is an nn.Module executing its vertices in
topological order on the MI355X op library.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Iterable, Optional, Sequence, Union

import torch
import torch.nn as nn

from ..ops import functional as OF
from ..ops.optim import Updater
from .layers import BaseLayer, OutputLayer


@dataclass
class InputType:
    """Input type declaration (DL4J InputType analog)."""

    kind: str               # "feedforward" | "convolutional_flat" | "convolutional"
    height: int = 0
    width: int = 0
    channels: int = 0
    size: int = 0

    @classmethod
    def feed_forward(cls, size: int) -> "InputType":
        return cls("feedforward", size=size)

    @classmethod
    def convolutional_flat(cls, h: int, w: int, c: int) -> "InputType":
        # reference Java:130-131: InputType.convolutionalFlat(28,28,1)
        return cls("convolutional_flat", height=h, width=w, channels=c)

    @classmethod
    def convolutional(cls, h: int, w: int, c: int) -> "InputType":
        return cls("convolutional", height=h, width=w, channels=c)

    def shape(self, n: int = 1) -> tuple:
        if self.kind == "feedforward":
            return (n, self.size)
        if self.kind == "convolutional_flat":
            return (n, self.height * self.width * self.channels)
        return (n, self.channels, self.height, self.width)


class GraphBuilder:
    """Builds a ComputationGraph.

    Defaults (seed, optimizer kind, clip, l2, default activation) mirror the
    reference's NeuralNetConfiguration.Builder chain (Java:118-128).
    """

    def __init__(
        self,
        seed: int = 666,
        default_activation: str = "identity",
        optim_cfg=None,
    ):
        from ..config import OptimConfig

        self.seed = seed
        self.default_activation = default_activation
        self.optim_cfg = optim_cfg or OptimConfig()
        self._inputs: list[str] = []
        self._input_types: dict[str, InputType] = {}
        self._outputs: list[str] = []
        self._vertices: dict[str, tuple[BaseLayer, list[str]]] = {}
        self._preprocessors: dict[str, BaseLayer] = {}

    def add_inputs(self, *names: str) -> "GraphBuilder":
        self._inputs.extend(names)
        return self

    def set_input_types(self, *types: InputType) -> "GraphBuilder":
        for name, t in zip(self._inputs, types):
            self._input_types[name] = t
        return self

    def add_layer(
        self, name: str, layer: BaseLayer, *inputs: str,
        preprocessor: Optional[BaseLayer] = None,
    ) -> "GraphBuilder":
        if name in self._vertices:
            raise KeyError(f"duplicate vertex {name!r}")
        layer.name = name
        self._vertices[name] = (layer, list(inputs))
        if preprocessor is not None:
            self._preprocessors[name] = preprocessor
        return self

    # DL4J spelling (inputPreProcessor("layer", proc))
    def input_preprocessor(self, layer_name: str, proc: BaseLayer) -> "GraphBuilder":
        self._preprocessors[layer_name] = proc
        return self

    def set_outputs(self, *names: str) -> "GraphBuilder":
        self._outputs = list(names)
        return self

    def build(self) -> "ComputationGraph":
        return ComputationGraph(
            inputs=self._inputs,
            input_types=self._input_types,
            outputs=self._outputs,
            vertices=self._vertices,
            preprocessors=self._preprocessors,
            seed=self.seed,
            optim_cfg=self.optim_cfg,
        )


class ComputationGraph(nn.Module):
    def __init__(self, inputs, input_types, outputs, vertices, preprocessors,
                 seed, optim_cfg):
        super().__init__()
        self.input_names = list(inputs)
        self.input_types = dict(input_types)
        self.output_names = list(outputs)
        self.seed = seed
        self.optim_cfg = optim_cfg
        self._topo = self._toposort(vertices)
        self._vertex_inputs = {k: v[1] for k, v in vertices.items()}
        self.layers = nn.ModuleDict({k: vertices[k][0] for k in self._topo})
        self.preprocessors = nn.ModuleDict(preprocessors)
        self._updater: Optional[Updater] = None
        self._initialized = False
        self._bn_fusion_pass()

    def _bn_fusion_pass(self) -> None:
        """Producer<->BatchNorm fusions for Conv/ConvT/Dense vertices whose
        SOLE consumer is a BatchNorm over the same feature axis (no
        preprocessor in between):

        - backward act-fusion (always on; kill switch GDLJ_NO_ACT_FUSE):
          BN's backward kernel applies the producer's activation backward
          and reduces the producer's bias gradient in the same pass,
          removing the standalone act_bwd_bias tensor streams.
        - statistics fusion (opt-in via GDLJ_BN_FUSE, measured ~neutral-
          to-negative on DCGAN-64): the producer's GPU epilogue also emits
          the BN batch statistics, skipping BN's own stats pass."""
        import os

        stats_fuse = os.environ.get("GDLJ_BN_FUSE", "0") == "1"
        from ..ops.gpu_ops import ACT_CODES
        from .layers import (BatchNormLayer, Conv2dLayer,
                             ConvTranspose2dLayer, DenseLayer, OutputLayer)

        consumers: dict[str, list[str]] = {}
        for name in self._topo:
            for src in self._vertex_inputs[name]:
                consumers.setdefault(src, []).append(name)
        for name in self._topo:
            layer = self.layers[name]
            if name in self.preprocessors:
                continue  # reshape between producer and consumer: axes differ
            srcs = self._vertex_inputs[name]
            if len(srcs) != 1 or srcs[0] not in self.layers:
                continue
            prod = self.layers[srcs[0]]
            if len(consumers.get(srcs[0], [])) != 1:
                continue
            if isinstance(prod, OutputLayer) or not isinstance(
                    prod, (Conv2dLayer, ConvTranspose2dLayer, DenseLayer)):
                continue
            prod_act = (prod.activation in ACT_CODES
                        and ACT_CODES[prod.activation] != 0)
            if isinstance(layer, BatchNormLayer):
                if stats_fuse:
                    prod.emit_bn_stats = True
                if prod_act:
                    layer.bwd_act = (ACT_CODES[prod.activation],
                                     getattr(prod, "slope", 0.2),
                                     getattr(prod, "bias", None) is not None)
            elif isinstance(layer, Conv2dLayer) and prod_act and \
                    not isinstance(prod, DenseLayer):
                # strided-conv consumer: its dgrad col2im folds the
                # producer's activation backward + bias grad in
                layer.prev_act = (ACT_CODES[prod.activation],
                                  getattr(prod, "slope", 0.2),
                                  getattr(prod, "bias", None) is not None)

    # ------------------------------------------------------------ build
    def _toposort(self, vertices) -> list[str]:
        order, seen, temp = [], set(), set()
        names = list(vertices)

        def visit(n):
            if n in seen or n in self.input_names:
                return
            if n in temp:
                raise ValueError(f"cycle at {n}")
            temp.add(n)
            for dep in vertices[n][1]:
                visit(dep)
            temp.discard(n)
            seen.add(n)
            order.append(n)

        for n in names:
            visit(n)
        return order

    def init(self) -> "ComputationGraph":
        """Allocate/initialize parameters (DL4J ComputationGraph.init())."""
        gen = torch.Generator().manual_seed(self.seed)
        for name in self._topo:
            layer = self.layers[name]
            if hasattr(layer, "reset_parameters"):
                layer.reset_parameters(gen)
        self._initialized = True
        return self

    # ---------------------------------------------------------- forward
    @staticmethod
    def _shape_input(t: Optional[InputType], x: torch.Tensor) -> torch.Tensor:
        if (t is not None and x.dim() == 2
                and t.kind in ("convolutional_flat", "convolutional")):
            return x.reshape(x.shape[0], t.channels, t.height, t.width)
        return x

    def forward(self, *inputs: torch.Tensor) -> torch.Tensor:
        acts: dict[str, torch.Tensor] = {}
        for name, x, in zip(self.input_names, inputs):
            acts[name] = self._shape_input(self.input_types.get(name), x)
        for name in self._topo:
            layer = self.layers[name]
            srcs = [acts[s] for s in self._vertex_inputs[name]]
            if name in self.preprocessors:
                srcs = [self.preprocessors[name](s) for s in srcs]
            acts[name] = layer(*srcs)
        outs = [acts[n] for n in self.output_names]
        return outs[0] if len(outs) == 1 else tuple(outs)

    @torch.no_grad()
    def feed_forward(self, *inputs: torch.Tensor, upto: Optional[str] = None):
        """Run the graph and return the activation of vertex `upto`
        (feature extraction; DL4J ComputationGraph.feedForward analog).
        With upto=None returns the dict of ALL vertex activations."""
        was_training = self.training
        self.eval()
        try:
            acts: dict[str, torch.Tensor] = {}
            for name, x in zip(self.input_names, inputs):
                acts[name] = self._shape_input(self.input_types.get(name), x)
            for name in self._topo:
                layer = self.layers[name]
                srcs = [acts[s] for s in self._vertex_inputs[name]]
                if name in self.preprocessors:
                    srcs = [self.preprocessors[name](s) for s in srcs]
                acts[name] = layer(*srcs)
                if name == upto:
                    return acts[name]
        finally:
            self.train(was_training)
        if upto is not None:
            raise KeyError(f"vertex {upto!r} not found")
        return acts

    @torch.no_grad()
    def output(self, *inputs: torch.Tensor) -> torch.Tensor:
        """Inference forward (reference gen.output(z), Java:420, 551):
        applies output-layer activations."""
        was_training = self.training
        self.eval()
        try:
            y = self.forward(*inputs)
        finally:
            self.train(was_training)
        outs = list(y) if isinstance(y, tuple) else [y]
        for i, name in enumerate(self.output_names):
            layer = self.layers[name]
            if isinstance(layer, OutputLayer):
                act = layer.inference_activation
                if act == "softmax":
                    outs[i] = torch.softmax(outs[i].float(), dim=1)
                else:
                    outs[i] = OF.activation(outs[i], act) if act != "identity" else outs[i]
        return outs[0] if len(outs) == 1 else tuple(outs)

    def compile_inference(self, *example_inputs: torch.Tensor):
        """Capture output() as a hipGraph for fixed-shape serving.

        Returns a callable f(*inputs) -> output that replays the captured
        graph (one launch per request instead of one per op). On CPU (or
        if capture fails) returns the plain output() path. Weights are
        read in-place at replay, so post-training weight updates are
        picked up as long as the packed-layout caches were rebuilt inside
        the capture (invalidate them before compiling after a weight
        change).
        """
        dev = next(self.parameters()).device
        if dev.type != "cuda":
            return lambda *xs: self.output(*xs)
        self.eval()
        statics = [x.clone().to(dev) for x in example_inputs]
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(2):
                    self.output(*statics)
            torch.cuda.current_stream().wait_stream(s)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                out = self.output(*statics)
        except Exception:  # pragma: no cover - capture unsupported
            return lambda *xs: self.output(*xs)

        def run(*xs: torch.Tensor) -> torch.Tensor:
            for st, x in zip(statics, xs):
                st.copy_(x.to(st.dtype), non_blocking=True)
            graph.replay()
            return out

        run.graph = graph  # keep the graph (and its pool) alive
        return run

    # ------------------------------------------------------------- loss
    def loss(self, logits, labels) -> torch.Tensor:
        """Loss of the (single) OutputLayer on logits."""
        out_layer = self.layers[self.output_names[0]]
        if not isinstance(out_layer, OutputLayer):
            raise TypeError("graph output vertex is not an OutputLayer")
        if out_layer.loss == "xent" and out_layer.inference_activation == "softmax":
            return OF.softmax_cross_entropy(logits, labels)
        return OF.LOSSES[out_layer.loss](logits, labels)

    # -------------------------------------------------------------- fit
    @property
    def updater(self) -> Updater:
        if self._updater is None:
            self._updater = Updater.for_graph(self, self.optim_cfg)
        return self._updater

    def fit(self, data, epochs: int = 1) -> float:
        """Train on DataSet(s)/iterator (SparkComputationGraph.fit analog).

        Returns the last minibatch loss.
        """
        from ..data.csv_reader import DataSet

        self.train()
        last = 0.0
        for _ in range(epochs):
            batches: Iterable
            if isinstance(data, DataSet):
                batches = [data]
            else:
                batches = data
            for ds in batches:
                dev = next(self.parameters()).device
                feats = ds.features.to(dev)
                labels = ds.labels.to(dev)
                self.updater.zero_grad()
                logits = self.forward(feats)
                loss = self.loss(logits, labels)
                loss.backward()
                self.updater.step()
                last = float(loss.detach())
        return last

    def fit_averaged(self, datasets) -> float:
        """ParameterAveragingTrainingMaster semantics for one fit call
        (Java:325-330, 425-426): each DataSet is an independent worker
        partition that starts from the CURRENT parameters and updater
        state, fits its one minibatch locally, and the resulting
        parameter vectors (and updater state — DL4J averages it by
        default) are averaged.  This is what
        `sparkDis.fit(sc.parallelize([real_ds, fake_ds]))` computes: ONE
        averaged update, not two sequential SGD steps.

        Returns the mean of the partition losses.
        """
        import copy as _copy

        datasets = list(datasets)
        if len(datasets) <= 1:
            return self.fit(datasets)
        init_p = [p.detach().clone() for p in self.parameters()]
        init_b = [b.detach().clone() for b in self.buffers()]
        init_u = _copy.deepcopy(self.updater.state_dict())
        acc_p = [torch.zeros_like(p, dtype=torch.float32) for p in init_p]
        acc_b = [torch.zeros_like(b, dtype=torch.float32) for b in init_b]
        acc_u: list = []
        losses = []
        for wi, ds in enumerate(datasets):
            if wi > 0:  # restore the broadcast state for this "worker"
                with torch.no_grad():
                    for p, p0 in zip(self.parameters(), init_p):
                        p.copy_(p0)
                    for b, b0 in zip(self.buffers(), init_b):
                        b.copy_(b0)
                self.updater.load_state_dict(_copy.deepcopy(init_u))
            losses.append(self.fit(ds))
            acc_u.append(_copy.deepcopy(self.updater.state_dict()))
            with torch.no_grad():
                for a, p in zip(acc_p, self.parameters()):
                    a += p.detach().float()
                for a, b in zip(acc_b, self.buffers()):
                    a += b.detach().float()
        n = float(len(datasets))
        with torch.no_grad():
            for p, a in zip(self.parameters(), acc_p):
                p.copy_((a / n).to(p.dtype))
            for b, a in zip(self.buffers(), acc_b):
                b.copy_((a / n).to(b.dtype))
        # average the updater slot tensors (m/v/master); scalars from the
        # last worker (t is identical across workers)
        avg = acc_u[-1]
        for key in ("m", "v", "master"):
            for si, slot in enumerate(avg["slots"]):
                if slot[key] is None:
                    continue
                slot[key] = sum(u["slots"][si][key].float()
                                for u in acc_u) / n
        self.updater.load_state_dict(avg)
        for layer in self.layers.values():
            for p in layer.parameters():
                if hasattr(p, "_gdlj_cache"):
                    del p._gdlj_cache
        return float(sum(losses) / n)

    # ----------------------------------------------------- param access
    def get_layer(self, name: str) -> BaseLayer:
        return self.layers[name]

    def layer_names(self) -> list[str]:
        return list(self._topo)

    def n_params(self) -> int:
        """Total param count incl. BN running stats (DL4J convention)."""
        return sum(p.numel() for p in self.parameters()) + sum(
            b.numel() for b in self.buffers()
        )

    def params_flat(self) -> torch.Tensor:
        """Flattened fp32 param vector in layer order (DL4J coefficients.bin
        ordering: per layer, W then b then gamma/beta/mean/var)."""
        chunks = []
        for name in self._topo:
            layer = self.layers[name]
            for key in ("W", "b", "gamma", "beta", "mean", "var"):
                if key in layer.param_keys():
                    chunks.append(layer.get_param(key).detach().float().reshape(-1).cpu())
        if not chunks:
            return torch.empty(0)
        return torch.cat(chunks)

    def load_params_flat(self, vec: torch.Tensor) -> None:
        off = 0
        for name in self._topo:
            layer = self.layers[name]
            for key in ("W", "b", "gamma", "beta", "mean", "var"):
                if key in layer.param_keys():
                    t = layer.get_param(key)
                    n = t.numel()
                    layer.set_param(key, vec[off : off + n].reshape(t.shape))
                    off += n
        if off != vec.numel():
            raise ValueError(f"param vector size mismatch: used {off}, got {vec.numel()}")

    # ---------------------------------------------------------- summary
    def summary(self, batch: int = 1) -> str:
        """Shape/param table (DL4J ComputationGraph.summary())."""
        lines = [f"{'name':<28}{'type':<26}{'out shape':<22}{'params':>10}"]
        lines.append("-" * 86)
        shapes: dict[str, tuple] = {}
        for name in self.input_names:
            t = self.input_types.get(name)
            if t is not None:
                s = t.shape(batch)
                if t.kind == "convolutional_flat":
                    s = (batch, t.channels, t.height, t.width)
                shapes[name] = s
            else:
                shapes[name] = (batch, -1)
        total = 0
        for name in self._topo:
            layer = self.layers[name]
            srcs = self._vertex_inputs[name]
            in_shapes = [shapes.get(s, (batch, -1)) for s in srcs]
            if name in self.preprocessors:
                in_shapes = [self.preprocessors[name].out_shape(s)
                             for s in in_shapes]
            # multi-input vertices (MergeVertex) take every input shape
            out_shape = layer.out_shape(*in_shapes)
            shapes[name] = out_shape
            n = layer.n_params()
            total += n
            lines.append(
                f"{name:<28}{type(layer).__name__:<26}{str(out_shape):<22}{n:>10}"
            )
        lines.append("-" * 86)
        lines.append(f"total params: {total}")
        return "\n".join(lines)

    def to_dot(self) -> str:
        """Graphviz DOT of the vertex topology (inputs, layers with
        param counts, preprocessor edges, outputs) — render with any
        `dot` tool; complements the tabular summary()."""
        lines = ["digraph ComputationGraph {", "  rankdir=TB;"]
        for name in self.input_names:
            lines.append(
                f'  "{name}" [shape=oval, style=filled, '
                f'fillcolor=lightblue];')
        for name in self._topo:
            layer = self.layers[name]
            n = layer.n_params()
            shape = ("doubleoctagon" if name in self.output_names
                     else "box")
            label = f"{name}\\n{type(layer).__name__}"
            if n:
                label += f"\\n{n:,} params"
            lines.append(f'  "{name}" [shape={shape}, label="{label}"];')
            for s in self._vertex_inputs[name]:
                edge = f'  "{s}" -> "{name}"'
                if name in self.preprocessors:
                    pname = type(self.preprocessors[name]).__name__
                    edge += f' [label="{pname}"]'
                lines.append(edge + ";")
        lines.append("}")
        return "\n".join(lines)

    # ------------------------------------------------------------ misc
    def clone(self) -> "ComputationGraph":
        import copy

        g = copy.deepcopy(self)
        g._updater = None
        return g

    def to_device(self, device, dtype=None) -> "ComputationGraph":
        self.to(device)
        if dtype is not None:
            for layer in self.layers.values():
                for p in layer.parameters(recurse=False):
                    # BN affine/stats stay fp32 (SURVEY hard-part #3)
                    from .layers import BatchNormLayer

                    if not isinstance(layer, BatchNormLayer):
                        p.data = p.data.to(dtype)
        return self
