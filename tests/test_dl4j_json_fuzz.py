"""Property-based DL4J-JSON round-trips over randomly composed graphs.

Complements the fixed-fixture tests in test_dl4j_json.py: hypothesis
composes random feed-forward graphs (Dense/BN/Activation chains, an
optional MergeVertex second input, an Output head with either loss)
from the emitter's supported layer set and asserts that
to_dl4j_json -> from_dl4j_json preserves topology, per-layer geometry
and — after copying the flat parameter vector — the forward outputs.
"""

import json

import pytest
import torch

try:
    from hypothesis import given, settings, strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.graph import (
    ActivationLayer,
    BatchNormLayer,
    DenseLayer,
    GraphBuilder,
    InputType,
    MergeVertex,
    OutputLayer,
)
from gan_deeplearning4j_amd.graph.dl4j_json import from_dl4j_json, to_dl4j_json

ACTS = ["identity", "tanh", "sigmoid", "relu", "lrelu"]


def _cfg():
    cfg = GanConfig()
    cfg.train.use_gpu = False
    return cfg


layer_spec = st.sampled_from(["dense", "bn", "act"])
chain = st.lists(
    st.tuples(layer_spec, st.integers(2, 24), st.sampled_from(ACTS)),
    min_size=1, max_size=5)


@settings(max_examples=25, deadline=None)
@given(
    chain=chain,
    n_in=st.integers(2, 16),
    merge_in=st.one_of(st.none(), st.integers(2, 8)),
    out_spec=st.tuples(st.integers(1, 10),
                       st.sampled_from([("sigmoid", "xent"),
                                        ("softmax", "mcxent")])),
    lr=st.sampled_from([0.002, 0.004, 0.0]),
)
def test_random_graph_roundtrip(chain, n_in, merge_in, out_spec, lr):
    cfg = _cfg()
    gb = GraphBuilder(seed=666, optim_cfg=cfg.optim)
    if merge_in is not None:
        gb.add_inputs("in_a", "in_b")
        gb.set_input_types(InputType.feed_forward(n_in),
                           InputType.feed_forward(merge_in))
        gb.add_layer("merge_0", MergeVertex(), "in_a", "in_b")
        prev, width = "merge_0", n_in + merge_in
    else:
        gb.add_inputs("in_a")
        gb.set_input_types(InputType.feed_forward(n_in))
        prev, width = "in_a", n_in
    for i, (kind, n, act) in enumerate(chain):
        name = f"l_{i}_{kind}"
        if kind == "dense":
            gb.add_layer(name, DenseLayer(width, n, act, lr=lr), prev)
            width = n
        elif kind == "bn":
            gb.add_layer(name, BatchNormLayer(width, lr=lr), prev)
        else:
            gb.add_layer(name, ActivationLayer(act), prev)
        prev = name
    n_out, (oact, loss) = out_spec
    gb.add_layer("out_0", OutputLayer(width, n_out, oact, loss, lr=lr),
                 prev)
    gb.set_outputs("out_0")
    g = gb.build().init()

    conf = json.loads(json.dumps(to_dl4j_json(g)))  # through real JSON
    g2 = from_dl4j_json(conf)
    assert g2.layer_names() == g.layer_names()
    assert g2.n_params() == g.n_params()
    g2.load_params_flat(g.params_flat())

    torch.manual_seed(0)
    xs = [torch.randn(3, n_in)]
    if merge_in is not None:
        xs.append(torch.randn(3, merge_in))
    g.eval()
    g2.eval()
    with torch.no_grad():
        y1 = g.output(*xs)
        y2 = g2.output(*xs)
    assert torch.allclose(y1, y2, atol=1e-6), (y1 - y2).abs().max()


@settings(max_examples=8, deadline=None)
@given(
    chain=chain,
    n_in=st.integers(2, 12),
    lr=st.sampled_from([0.002, 0.0]),
)
def test_random_graph_zip_roundtrip(chain, n_in, lr):
    # full ModelSerializer zip round trip (DL4J layout + coefficients +
    # updater state) over the same random graph family
    import tempfile
    from pathlib import Path

    from gan_deeplearning4j_amd.graph.serialization import ModelSerializer

    cfg = _cfg()
    gb = GraphBuilder(seed=666, optim_cfg=cfg.optim)
    gb.add_inputs("in_a")
    gb.set_input_types(InputType.feed_forward(n_in))
    prev, width = "in_a", n_in
    for i, (kind, n, act) in enumerate(chain):
        name = f"l_{i}_{kind}"
        if kind == "dense":
            gb.add_layer(name, DenseLayer(width, n, act, lr=lr), prev)
            width = n
        elif kind == "bn":
            gb.add_layer(name, BatchNormLayer(width, lr=lr), prev)
        else:
            gb.add_layer(name, ActivationLayer(act), prev)
        prev = name
    gb.add_layer("out_0", OutputLayer(width, 3, "softmax", "mcxent", lr=lr),
                 prev)
    gb.set_outputs("out_0")
    g = gb.build().init()

    with tempfile.TemporaryDirectory() as td:
        p = Path(td) / "m.zip"
        ModelSerializer.write_model(g, p, save_updater=True)
        g2 = ModelSerializer.restore_computation_graph(p)
    assert g2.layer_names() == g.layer_names()
    torch.manual_seed(1)
    x = torch.randn(3, n_in)
    g.eval()
    g2.eval()
    with torch.no_grad():
        assert torch.allclose(g.output(x), g2.output(x), atol=1e-6)
