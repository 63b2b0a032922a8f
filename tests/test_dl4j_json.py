"""DL4J-beta3-style configuration.json round-trips.

Byte-compat against a real DL4J install is unvalidated offline (no JVM);
these tests pin what CAN be validated here: the emitter produces the
beta3 @class-tagged structure, the parser rebuilds an equivalent graph
from it (including a hand-constructed known-good fixture written the way
DL4J's Jackson emits it), and a zip stripped to the DL4J-style
configuration.json alone still restores with coefficients applied.
"""

import json
import zipfile

import pytest
import torch

from gan_deeplearning4j_amd.config import GanConfig
from gan_deeplearning4j_amd.graph.dl4j_json import from_dl4j_json, to_dl4j_json
from gan_deeplearning4j_amd.graph.serialization import ModelSerializer
from gan_deeplearning4j_amd.models.reference_protocol import (
    build_discriminator,
    build_frozen_generator,
    build_stacked_gan,
    build_transfer_classifier,
)


def _cfg():
    cfg = GanConfig()
    cfg.train.use_gpu = False
    cfg.optim.optimizer = "rmsprop"
    cfg.optim.rms_decay = 1e-8
    return cfg


@pytest.mark.parametrize("builder", ["dis", "gen", "gan", "cv"])
def test_dl4j_json_roundtrip_reference_graphs(builder):
    cfg = _cfg()
    dis = build_discriminator(cfg)
    g = {
        "dis": dis,
        "gen": build_frozen_generator(cfg),
        "gan": build_stacked_gan(cfg),
        "cv": build_transfer_classifier(dis, cfg),
    }[builder]
    conf = to_dl4j_json(g)
    # beta3 structural invariants
    assert conf["networkInputs"] and conf["networkOutputs"]
    for name, v in conf["vertices"].items():
        assert v["@class"].startswith("org.deeplearning4j.nn.conf.graph.")
        if v["@class"].endswith("LayerVertex"):
            assert v["layerConf"]["layer"]["@class"].startswith(
                "org.deeplearning4j.nn.conf.layers.")
            assert v["layerConf"]["layer"]["layerName"] == name
    g2 = from_dl4j_json(json.loads(json.dumps(conf)))
    assert g2.layer_names() == g.layer_names()
    assert g2.n_params() == g.n_params()
    # same per-layer geometry: flat param vectors have identical length
    assert g2.params_flat().numel() == g.params_flat().numel()
    # updater constants recovered
    assert g2.optim_cfg.optimizer == "rmsprop"
    assert g2.optim_cfg.rms_decay == 1e-8


def test_dl4j_style_zip_restores_without_native_sidecar(tmp_path):
    """Simulate a DL4J-written zip: only configuration.json (DL4J style)
    + coefficients.bin. Params must land in the same layer order."""
    cfg = _cfg()
    g = build_discriminator(cfg)
    p = tmp_path / "dis.zip"
    ModelSerializer.write_model(g, p, save_updater=False)
    # strip to the DL4J-visible members
    stripped = tmp_path / "dis_dl4j_only.zip"
    with zipfile.ZipFile(p) as zin, zipfile.ZipFile(stripped, "w") as zout:
        for name in ("configuration.json", "coefficients.bin"):
            zout.writestr(name, zin.read(name))
    g2 = ModelSerializer.restore_computation_graph(stripped)
    assert torch.allclose(g2.params_flat(), g.params_flat())
    x = torch.rand(4, 784)
    assert torch.allclose(g2.output(x), g.output(x), atol=1e-5)


# A hand-constructed configuration.json written the way DL4J beta3's
# Jackson emits a 2-vertex graph (Dense -> Output) — the known-good
# fixture the parser must accept (field order/extras included).
KNOWN_GOOD = {
    "backpropType": "Standard",
    "cacheMode": "NONE",
    "epochCount": 0,
    "inferenceWorkspaceMode": "ENABLED",
    "iterationCount": 0,
    "networkInputs": ["in"],
    "networkInputTypes": [{
        "@class": "org.deeplearning4j.nn.conf.inputs."
                  "InputType$InputTypeFeedForward",
        "size": 8,
    }],
    "networkOutputs": ["out"],
    "tbpttBackLength": 20,
    "tbpttFwdLength": 20,
    "trainingWorkspaceMode": "ENABLED",
    "validateOutputLayerConfig": True,
    "vertexInputs": {"hidden": ["in"], "out": ["hidden"]},
    "vertices": {
        "hidden": {
            "@class": "org.deeplearning4j.nn.conf.graph.LayerVertex",
            "layerConf": {
                "cacheMode": "NONE",
                "layer": {
                    "@class":
                        "org.deeplearning4j.nn.conf.layers.DenseLayer",
                    "activationFn": {
                        "@class": "org.nd4j.linalg.activations.impl."
                                  "ActivationTanH"},
                    "biasInit": 0.0,
                    "constraints": None,
                    "gradientNormalization":
                        "ClipElementWiseAbsoluteValue",
                    "gradientNormalizationThreshold": 1.0,
                    "hasBias": True,
                    "iupdater": {
                        "@class":
                            "org.nd4j.linalg.learning.config.RmsProp",
                        "epsilon": 1e-08,
                        "learningRate": 0.002,
                        "rmsDecay": 1e-08,
                    },
                    "l1": 0.0, "l1Bias": 0.0, "l2": 1e-4, "l2Bias": 0.0,
                    "layerName": "hidden",
                    "nin": 8, "nout": 4,
                    "weightInit": "XAVIER",
                },
                "miniBatch": True,
                "minimize": True,
                "optimizationAlgo": "STOCHASTIC_GRADIENT_DESCENT",
                "seed": 666,
                "variables": ["W", "b"],
            },
            "preProcessor": None,
        },
        "out": {
            "@class": "org.deeplearning4j.nn.conf.graph.LayerVertex",
            "layerConf": {
                "layer": {
                    "@class":
                        "org.deeplearning4j.nn.conf.layers.OutputLayer",
                    "activationFn": {
                        "@class": "org.nd4j.linalg.activations.impl."
                                  "ActivationSigmoid"},
                    "hasBias": True,
                    "iupdater": {
                        "@class":
                            "org.nd4j.linalg.learning.config.RmsProp",
                        "epsilon": 1e-08,
                        "learningRate": 0.002,
                        "rmsDecay": 1e-08,
                    },
                    "layerName": "out",
                    "lossFn": {"@class": "org.nd4j.linalg.lossfunctions."
                                         "impl.LossBinaryXENT"},
                    "nin": 4, "nout": 1,
                },
                "seed": 666,
            },
            "preProcessor": None,
        },
    },
}


def test_parse_hand_constructed_dl4j_fixture():
    g = from_dl4j_json(json.loads(json.dumps(KNOWN_GOOD)))
    assert g.layer_names() == ["hidden", "out"]
    h = g.get_layer("hidden")
    assert (h.n_in, h.n_out, h.activation) == (8, 4, "tanh")
    o = g.get_layer("out")
    assert (o.n_in, o.n_out, o.loss) == (4, 1, "xent")
    assert g.optim_cfg.optimizer == "rmsprop"
    assert g.optim_cfg.rms_decay == 1e-8
    y = g.output(torch.rand(3, 8))
    assert y.shape == (3, 1)


def test_dl4j_parser_rejects_unknown_layer():
    bad = json.loads(json.dumps(KNOWN_GOOD))
    bad["vertices"]["hidden"]["layerConf"]["layer"]["@class"] = (
        "org.deeplearning4j.nn.conf.layers.LocallyConnected2D")
    with pytest.raises(TypeError, match="LocallyConnected2D"):
        from_dl4j_json(bad)


def test_dl4j_parser_rejects_dangling_wiring():
    bad = json.loads(json.dumps(KNOWN_GOOD))
    bad["vertexInputs"]["hidden"] = ["missing_vertex"]
    with pytest.raises(ValueError, match="cyclic or dangling"):
        from_dl4j_json(bad)
