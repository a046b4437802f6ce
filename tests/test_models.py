"""Model-zoo smoke/shape tests on CPU (tiny configs)."""

import os
import subprocess
import sys

import pytest
import torch

from adapcc_amd.models.gpt2 import GPT2, GPT2Config
from adapcc_amd.models.resnet import ResNet18
from adapcc_amd.models.vgg import VGG16
from adapcc_amd.models.vit import ViT, ViTConfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_gpt2_tiny_forward_backward():
    cfg = GPT2Config.tiny()
    m = GPT2(cfg)
    x = torch.randint(0, cfg.vocab_size, (2, 32))
    logits, loss = m(x, x)
    assert logits.shape == (2, 32, cfg.vocab_size)
    loss.backward()
    assert m.wte.weight.grad is not None


def test_gpt2_small_param_count():
    m = GPT2(GPT2Config.small())
    n = m.num_params()
    assert 120e6 < n < 130e6, n  # GPT-2 small is ~124M


def test_vit_tiny():
    cfg = ViTConfig.tiny()
    m = ViT(cfg)
    x = torch.randn(2, 3, cfg.image_size, cfg.image_size)
    y = m(x)
    assert y.shape == (2, cfg.num_classes)
    y.sum().backward()


def test_vgg16_small_input():
    m = VGG16(num_classes=10, in_size=64)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)
    y.sum().backward()


def test_resnet18():
    m = ResNet18(num_classes=10)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 10)
    y.sum().backward()


def test_gns_estimators():
    from adapcc_amd.utils.gns import GNS

    torch.manual_seed(0)
    g = torch.randn(1000)
    noise = torch.randn(1000)
    g_small = g + noise          # batch 1: variance sigma^2
    g_big = g + noise * 0.5      # batch 4: variance sigma^2/4
    g2, tr = GNS.estimate_pair(g_small, g_big, 1, 4)
    assert tr > 0
    gns = GNS().compute_gns(g_small, g_big, 1, 4)
    assert gns == pytest.approx(tr / g2)


def test_elastic_state_roundtrip(tmp_path):
    sys.path.insert(0, os.path.join(REPO, "examples"))
    from train_elastic import State

    m = ResNet18(num_classes=10)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    st = State(m, opt)
    st.step = 7
    path = str(tmp_path / "ck.pt")
    st.save(path)

    m2 = ResNet18(num_classes=10)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1)
    st2 = State(m2, opt2)
    st2.load_and_sync(path, torch.device("cpu"))
    assert st2.step == 7
    for a, b in zip(m.parameters(), m2.parameters()):
        assert torch.equal(a, b)


@pytest.mark.parametrize("script,extra", [
    ("train_ddp.py", ["--steps", "2", "--batch", "2", "--image_size", "64"]),
    ("examples/train_moe.py", ["--steps", "2", "--batch", "2", "--seq", "32",
                               "--d_model", "64", "--d_hidden", "128"]),
])
def test_template_scripts_2proc_cpu(script, extra):
    import random

    port = random.randint(20000, 40000)
    env = dict(os.environ, ADAPCC_TRANSPORT="pg")
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port), os.path.join(REPO, script)] + extra
    res = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         env=env, cwd=REPO)
    assert res.returncode == 0, res.stdout[-2000:] + res.stderr[-2000:]
    assert "loss" in res.stdout


def test_fused_ln_autocast_fallback_cpu():
    """Under autocast (mixed weight/input dtypes) FusedLayerNorm must take
    the stock fallback and match nn.LayerNorm exactly."""
    from adapcc_amd.ops.fused import FusedLayerNorm

    torch.manual_seed(0)
    fused = FusedLayerNorm(64)
    ref = torch.nn.LayerNorm(64)
    with torch.no_grad():
        ref.weight.copy_(fused.weight)
        ref.bias.copy_(fused.bias)
    x = torch.randn(4, 64)
    with torch.autocast("cpu", dtype=torch.bfloat16):
        y1 = fused(x)
        y2 = ref(x)
    assert y1.dtype == y2.dtype
    assert torch.allclose(y1.float(), y2.float())
