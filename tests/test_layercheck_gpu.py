"""Per-layer numerics gate (VERDICT round-1 item 6): replaces the loose
whole-model e2e bounds as the kernel regression detector. Every node
the GPU plan materializes is compared against the CPU fp32 executor on
the same IR; per-op-class budgets scaled by sqrt(depth)."""
import numpy as np
import pytest
import torch

from tensorflow_web_deploy_amd.conf.config import Config
from tensorflow_web_deploy_amd.engine.layercheck import per_layer_compare
from tensorflow_web_deploy_amd.engine.service import build_graph

pytestmark = pytest.mark.gpu


def _check(model, size, chans, ncls, batch=4):
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    cfg = Config(model_name=model, device="cuda", input_height=size,
                 input_width=size, input_channels=chans,
                 num_classes=ncls)
    g = build_graph(cfg)
    rng = np.random.default_rng(11)
    x = rng.random((batch, size, size, chans), dtype=np.float32)
    rep = per_layer_compare(g, cfg, x)
    assert rep.rows, "nothing compared"
    assert not rep.violations(), "\n" + rep.summary()
    print("\n" + rep.summary())


@pytest.mark.timeout(900)
def test_layers_resnet50():
    _check("resnet50", 224, 3, 1000)


@pytest.mark.timeout(900)
def test_layers_inception_v3():
    _check("inception_v3", 299, 3, 1000)


@pytest.mark.timeout(900)
def test_layers_vit_b16():
    _check("vit_b16", 224, 3, 1000)


@pytest.mark.timeout(900)
def test_layers_mobilenet_v2():
    _check("mobilenet_v2", 224, 3, 1000)


@pytest.mark.timeout(300)
def test_layers_lenet():
    _check("lenet", 28, 1, 10, batch=8)


@pytest.mark.timeout(900)
def test_layers_vgg16():
    _check("vgg16", 224, 3, 1000, batch=2)


@pytest.mark.timeout(900)
def test_layers_densenet121():
    _check("densenet121", 224, 3, 1000, batch=2)
