"""Model-family tests: shapes, frozen-base semantics, save/load round trip."""
import torch

from ddlw_amd.core import tracking
from ddlw_amd.core.model_io import load_model, log_model
from ddlw_amd.models import build_model, build_resnet50, build_small_cnn


def test_small_cnn_shapes():
    m = build_small_cnn(num_classes=5)
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 5)


def test_resnet50_shapes():
    m = build_resnet50(num_classes=7)
    y = m(torch.randn(2, 3, 224, 224))
    assert y.shape == (2, 7)
    n_params = sum(p.numel() for p in m.parameters())
    # ResNet-50 @1000 classes has 25.56M; at 7 classes ~23.5-24M
    assert 20e6 < n_params < 30e6


def test_transfer_model_frozen_base():
    m = build_model(num_classes=5, dropout=0.5)
    head_params = [p for p in m.parameters() if p.requires_grad]
    # only the Dense head trains (reference P1/02:167-176)
    assert sum(p.numel() for p in head_params) == 1280 * 5 + 5
    m.train()
    # frozen BN stays in eval mode (quirk 5)
    assert not m.base.base.training
    y = m(torch.randn(2, 3, 64, 64))
    assert y.shape == (2, 5)
    # backward only touches the head
    y.sum().backward()
    assert m.classifier.weight.grad is not None


def test_model_save_load_roundtrip(ddlw_home):
    tracking.set_experiment("models")
    m = build_small_cnn(num_classes=4)
    with tracking.start_run() as run:
        uri = log_model(m, "model")
    m2 = load_model(uri)
    x = torch.randn(3, 3, 64, 64)
    m.eval(), m2.eval()
    assert torch.allclose(m(x), m2(x), atol=1e-6)
    # also via the registry stage URI
    tracking.register_model(uri, "cnn")
    tracking.transition_model_version_stage("cnn", 1, "Production")
    m3 = load_model("models:/cnn/production")
    m3.eval()
    assert torch.allclose(m(x), m3(x), atol=1e-6)


def test_fused_block_gating_cpu(monkeypatch):
    """bottleneck_fusable must refuse CPU/eval/disabled configurations
    (the fused Function is a GPU-training-only path)."""
    import torch

    from ddlw_amd.models.resnet import Bottleneck
    from ddlw_amd.ops import block as B

    blk = Bottleneck(256, 64)
    x = torch.randn(2, 256, 14, 14)
    blk.train()
    assert not B.bottleneck_fusable(blk, x)  # CPU tensor
    monkeypatch.setenv("DDLW_FUSED_BLOCK", "0")
    assert not B.bottleneck_fusable(blk, x)  # disabled
    monkeypatch.delenv("DDLW_FUSED_BLOCK")
    blk.eval()
    assert not B.bottleneck_fusable(blk, x)  # eval mode
    blk.train()
    with torch.no_grad():
        assert not B.bottleneck_fusable(blk, x)  # grad disabled


def test_conv_backward_helper_cpu_fallback_with_acc():
    """The shared conv_backward helper's library fallback must apply the
    accumulate input (the fused block's join-add) identically to autograd."""
    import torch
    import torch.nn.functional as F

    from ddlw_amd.ops.conv_gemm import conv_backward

    torch.manual_seed(0)
    x = torch.randn(2, 4, 8, 8, requires_grad=True)
    w = torch.randn(6, 4, 3, 3, requires_grad=True)
    acc = torch.randn(2, 4, 8, 8)
    y = F.conv2d(x, w, None, 1, 1)
    dy = torch.randn_like(y)
    y.backward(dy)
    dx, dw = conv_backward(dy, x.detach(), w.detach(), 1, 1,
                           hip_dgrad=False, hip_wgrad=False, acc=acc)
    assert torch.allclose(dx.float(), (x.grad + acc).float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(dw.float(), w.grad.float(), atol=2e-2, rtol=2e-2)
