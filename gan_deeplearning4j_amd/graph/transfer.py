"""Transfer-learning API (DL4J TransferLearning.GraphBuilder analog).

Recreates the reference's classifier construction (Java:337-364):
  new TransferLearning.GraphBuilder(dis)
      .fineTuneConfiguration(...)             # re-apply optimizer defaults
      .setFeatureExtractor("dis_dense_layer_6")  # freeze through this layer
      .removeVertexKeepConnections("dis_output_layer_7")
      .addLayer(...)                          # new BN + softmax head
      .build()
Returns a NEW ComputationGraph with copied weights.
"""

from __future__ import annotations

import copy
from dataclasses import dataclass, field
from typing import Optional

from .builder import ComputationGraph, GraphBuilder
from .layers import BaseLayer


@dataclass
class FineTuneConfiguration:
    """Optimizer defaults re-applied to the transferred graph
    (reference Java:338-349)."""

    optim_cfg: object = None
    seed: int = 666


class TransferLearningBuilder:
    def __init__(self, source: ComputationGraph):
        self.source = source
        self._fine_tune: Optional[FineTuneConfiguration] = None
        self._freeze_through: Optional[str] = None
        self._removed: list[str] = []
        self._added: list[tuple[str, BaseLayer, tuple[str, ...]]] = []
        self._new_outputs: Optional[list[str]] = None

    def fine_tune_configuration(self, cfg: FineTuneConfiguration):
        self._fine_tune = cfg
        return self

    def set_feature_extractor(self, layer_name: str):
        """Freeze all vertices up to and including layer_name (Java:350)."""
        self._freeze_through = layer_name
        return self

    def remove_vertex_keep_connections(self, name: str):
        self._removed.append(name)
        return self

    def add_layer(self, name: str, layer: BaseLayer, *inputs: str):
        self._added.append((name, layer, inputs))
        return self

    def set_outputs(self, *names: str):
        self._new_outputs = list(names)
        return self

    def build(self) -> ComputationGraph:
        src = self.source
        optim_cfg = (
            self._fine_tune.optim_cfg
            if self._fine_tune and self._fine_tune.optim_cfg
            else src.optim_cfg
        )
        seed = self._fine_tune.seed if self._fine_tune else src.seed
        gb = GraphBuilder(seed=seed, optim_cfg=optim_cfg)
        gb.add_inputs(*src.input_names)
        gb.set_input_types(*[src.input_types[n] for n in src.input_names
                             if n in src.input_types])

        # Copy surviving vertices (deep-copied layers => copied weights).
        removed = set(self._removed)
        reconnect: dict[str, list[str]] = {}
        for name in src.layer_names():
            inputs = list(src._vertex_inputs[name])
            # rewire inputs that point at removed vertices to THEIR inputs
            fixed = []
            for s in inputs:
                while s in removed:
                    s = src._vertex_inputs[s][0]
                fixed.append(s)
            if name in removed:
                reconnect[name] = fixed
                continue
            layer = copy.deepcopy(src.layers[name])
            proc = (
                copy.deepcopy(src.preprocessors[name])
                if name in src.preprocessors
                else None
            )
            gb.add_layer(name, layer, *fixed, preprocessor=proc)

        # Added layers: inputs referencing removed vertices rewired likewise.
        last_added = None
        for name, layer, inputs in self._added:
            fixed = []
            for s in inputs:
                while s in removed:
                    s = reconnect[s][0] if s in reconnect else src._vertex_inputs[s][0]
                fixed.append(s)
            gb.add_layer(name, layer, *fixed)
            last_added = name

        outputs = self._new_outputs
        if outputs is None:
            outputs = [o if o not in removed else last_added for o in src.output_names]
            outputs = [o for o in outputs if o is not None]
        gb.set_outputs(*outputs)

        g = gb.build()
        # init only the newly added layers (copied ones keep weights)
        import torch

        gen = torch.Generator().manual_seed(seed)
        new_names = {n for n, _, _ in self._added}
        for name in g.layer_names():
            layer = g.layers[name]
            if name in new_names and hasattr(layer, "reset_parameters"):
                layer.reset_parameters(gen)
        g._initialized = True

        # Freeze through the feature-extractor boundary (topological order).
        if self._freeze_through is not None:
            for name in g.layer_names():
                g.layers[name].set_frozen(True)
                if name == self._freeze_through:
                    break
        return g
