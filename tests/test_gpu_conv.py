"""Hand-written MFMA conv kernels vs PyTorch reference (fp32, on
bf16-rounded operands so only accumulation-order noise remains)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"

SHAPES = [
    # (H, W, C, KH, KW, S, COUT)  — Ape-X/R2D2 + IMPALA stacks
    (84, 84, 4, 8, 8, 4, 32),
    (20, 20, 32, 4, 4, 2, 64),
    (9, 9, 64, 3, 3, 1, 64),
    (84, 84, 4, 8, 8, 4, 16),
    (20, 20, 16, 4, 4, 2, 32),
]


@pytest.fixture(scope="module")
def ext():
    from distributed_rl_amd.ops import hip_ext

    return hip_ext(required=True)


def test_mfma_fragment_map(ext):
    torch.manual_seed(0)
    A = torch.randn(16, 32, device=DEV).to(torch.bfloat16)
    # asymmetric B catches transposed C-writes (guide §3)
    B = (torch.arange(32 * 16, device=DEV).reshape(32, 16) % 7
         ).float().to(torch.bfloat16) * 0.1 + torch.randn(32, 16, device=DEV
                                                          ).to(torch.bfloat16) * 0.01
    C = torch.zeros(16, 16, device=DEV)
    ext.mfma_probe(A.contiguous(), B.contiguous(), C)
    torch.cuda.synchronize()
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=1e-2, rtol=1e-2), (C - ref).abs().max()


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_fwd_bf16_vs_aten(ext, shape):
    from distributed_rl_amd import ops

    H, W, C, KH, KW, S, COUT = shape
    torch.manual_seed(1)
    N = 33
    x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    w = (torch.randn(COUT, C, KH, KW, device=DEV) * 0.05).to(
        torch.bfloat16).to(memory_format=torch.channels_last)
    b = (torch.randn(COUT, device=DEV) * 0.1).to(torch.bfloat16)
    out = ops.fused_conv_relu(x, w, b, S)
    ref = F.relu(F.conv2d(x.float(), w.float(), b.float(), stride=S))
    assert out.shape == ref.shape
    err = (out.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 2e-2, f"{shape}: max err {err} scale {scale}"


def test_conv_fwd_u8_fused_dequant(ext):
    from distributed_rl_amd import ops

    torch.manual_seed(2)
    N, C, H, W, COUT, S = 17, 4, 84, 84, 32, 4
    x = torch.randint(0, 256, (N, C, H, W), dtype=torch.uint8, device=DEV).to(
        memory_format=torch.channels_last)
    w = (torch.randn(COUT, C, 8, 8, device=DEV) * 0.05).to(
        torch.bfloat16).to(memory_format=torch.channels_last)
    b = torch.zeros(COUT, device=DEV, dtype=torch.bfloat16)
    out = ops.fused_conv_relu(x, w, b, S)
    xf = (x.float() / 255.0).to(torch.bfloat16).float()
    ref = F.relu(F.conv2d(xf, w.float(), None, stride=S))
    err = (out.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    assert err / scale < 2e-2, f"max err {err} scale {scale}"


def test_conv_backward_matches_eager(ext):
    from distributed_rl_amd import ops

    torch.manual_seed(3)
    N, C, H, W, COUT, S = 8, 32, 20, 20, 64, 2
    x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    w = ((torch.randn(COUT, C, 4, 4, device=DEV) * 0.05).to(torch.bfloat16)
         .to(memory_format=torch.channels_last).requires_grad_(True))
    b = (torch.randn(COUT, device=DEV) * 0.1).to(torch.bfloat16).requires_grad_(True)
    out = ops.fused_conv_relu(x, w, b, S)
    g = torch.randn_like(out)
    out.backward(g)
    gx, gw, gb = x.grad.clone(), w.grad.clone(), b.grad.clone()

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    ref = F.relu(F.conv2d(x2, w2, b2, stride=S))
    ref.backward(g)
    # The ReLU mask can flip on entries where |out| ~ bf16 noise, which puts
    # full |g| error on a handful of elements — compare statistically: small
    # mean error, and <1% of entries off by more than 10% of scale.
    for mine, theirs, name in [(gx, x2.grad, "gx"), (gw, w2.grad, "gw"),
                               (gb, b2.grad, "gb")]:
        m, t = mine.float(), theirs.float()
        scale = t.abs().max().item() + 1e-6
        mae = (m - t).abs().mean().item()
        frac_big = ((m - t).abs() > 0.1 * scale).float().mean().item()
        assert mae / scale < 5e-3, f"{name}: mae {mae} scale {scale}"
        assert frac_big < 0.01, f"{name}: {frac_big:.3%} entries off"


def test_model_fused_path_active(ext):
    """BaseAgent CNN2D must route through the fused kernels for bf16
    channels_last input on GPU (and match the eager path numerically)."""
    from distributed_rl_amd.config import load_config
    from distributed_rl_amd.models import BaseAgent

    net = BaseAgent(load_config("ape_x").model_info).to(DEV)
    net = net.to(memory_format=torch.channels_last).to(torch.bfloat16)
    cnn = net.nodes["module00"]
    x = torch.randn(4, 4, 84, 84, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    assert cnn._fused_ok(x)
    y_fused = cnn(x)
    cnn._fused_checked[(tuple(x.shape[1:]), x.dtype)] = False
    y_eager = cnn(x)
    cnn._fused_checked.clear()
    err = (y_fused.float() - y_eager.float()).abs().max().item()
    scale = y_eager.float().abs().max().item() + 1e-6
    assert err / scale < 3e-2


@pytest.mark.parametrize("shape", SHAPES)
def test_conv_wrw_vs_aten(ext, shape):
    H, W, C, KH, KW, S, COUT = shape
    torch.manual_seed(5)
    N = 37
    P, Q = (H - KH) // S + 1, (W - KW) // S + 1
    x = torch.randn(N, C, H, W, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    gout = torch.randn(N, COUT, P, Q, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    ws = torch.zeros(COUT, KH * KW * C, dtype=torch.float32, device=DEV)
    gb_ws = torch.zeros(COUT, dtype=torch.float32, device=DEV)
    ext.conv_wrw(x, gout, ws, gb_ws, S)
    torch.cuda.synchronize()
    gw = ws.view(COUT, KH, KW, C).permute(0, 3, 1, 2)
    gb_ref = gout.float().sum(dim=(0, 2, 3))
    assert torch.allclose(gb_ws, gb_ref, rtol=2e-2, atol=2e-2)
    _, gw_ref, _ = torch.ops.aten.convolution_backward(
        gout, x, torch.empty(COUT, C, KH, KW, device=DEV, dtype=torch.bfloat16
                             ).to(memory_format=torch.channels_last),
        None, [S, S], [0, 0], [1, 1], False, [0, 0], 1, [False, True, False])
    scale = gw_ref.float().abs().max().item() + 1e-6
    err = (gw - gw_ref.float()).abs().max().item()
    assert err / scale < 2e-2, f"{shape}: {err} vs {scale}"


def test_conv_wrw_u8_input(ext):
    torch.manual_seed(6)
    N, C, H, W, COUT, S, KH = 21, 4, 84, 84, 32, 4, 8
    P = Q = 20
    x = torch.randint(0, 256, (N, C, H, W), dtype=torch.uint8, device=DEV).to(
        memory_format=torch.channels_last)
    gout = torch.randn(N, COUT, P, Q, device=DEV).to(torch.bfloat16).to(
        memory_format=torch.channels_last)
    ws = torch.zeros(COUT, KH * KH * C, dtype=torch.float32, device=DEV)
    ext.conv_wrw(x, gout, ws, torch.empty(0, device=DEV), S)
    torch.cuda.synchronize()
    gw = ws.view(COUT, KH, KH, C).permute(0, 3, 1, 2)
    xf = (x.float() / 255.0).to(torch.bfloat16)
    _, gw_ref, _ = torch.ops.aten.convolution_backward(
        gout, xf, torch.empty(COUT, C, KH, KH, device=DEV, dtype=torch.bfloat16
                              ).to(memory_format=torch.channels_last),
        None, [S, S], [0, 0], [1, 1], False, [0, 0], 1, [False, True, False])
    scale = gw_ref.float().abs().max().item() + 1e-6
    err = (gw - gw_ref.float()).abs().max().item()
    assert err / scale < 2e-2, f"{err} vs {scale}"


DGRAD_SHAPES = [
    (20, 20, 32, 4, 4, 2, 64),
    (9, 9, 64, 3, 3, 1, 64),
    (20, 20, 16, 4, 4, 2, 32),
]


@pytest.mark.parametrize("shape", DGRAD_SHAPES)
def test_conv_dgrad_vs_aten(ext, shape):
    """Hand-written dgrad (masked-tap gather vs transposed weights) against
    aten convolution_backward on identical bf16 operands."""
    H, W, C, KH, KW, S, COUT = shape
    torch.manual_seed(5)
    N = 32
    P, Q = (H - KH) // S + 1, (W - KW) // S + 1
    gout = (torch.randn(N, COUT, P, Q, device=DEV) * 0.5).to(
        torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = (torch.randn(COUT, C, KH, KW, device=DEV) * 0.1).to(
        torch.bfloat16).contiguous(memory_format=torch.channels_last)
    dx = torch.empty(N, C, H, W, dtype=torch.bfloat16, device=DEV
                     ).contiguous(memory_format=torch.channels_last)
    w_t = torch.empty(w.numel(), dtype=torch.bfloat16, device=DEV)
    ext.conv_dgrad(gout, w, w_t, dx, S)
    torch.cuda.synchronize()
    x_dummy = torch.zeros(N, C, H, W, device=DEV, dtype=torch.bfloat16
                          ).contiguous(memory_format=torch.channels_last)
    ref, _, _ = torch.ops.aten.convolution_backward(
        gout.float(), x_dummy.float(), w.float(), None, [S, S], [0, 0],
        [1, 1], False, [0, 0], 1, [True, False, False])
    assert torch.allclose(dx.float(), ref, atol=3e-2, rtol=3e-2), \
        (dx.float() - ref).abs().max()


def test_conv_dgrad_in_fused_backward(ext, monkeypatch):
    """End-to-end: the fused conv autograd must produce the same input grad
    with the hand-written dgrad as aten does on the fp32 path."""
    from distributed_rl_amd.ops import fused_conv_relu

    monkeypatch.setenv("DRL_OWN_DGRAD", "1")

    torch.manual_seed(6)
    N, C, H, W, COUT, KH, S = 16, 32, 20, 20, 64, 4, 2
    x = (torch.randn(N, C, H, W, device=DEV) * 0.5).to(torch.bfloat16
        ).contiguous(memory_format=torch.channels_last).requires_grad_(True)
    w = ((torch.randn(COUT, C, KH, KH, device=DEV) * 0.05).to(torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_(True))
    b = torch.randn(COUT, device=DEV).to(torch.bfloat16).requires_grad_(True)
    out = fused_conv_relu(x, w, b, S)
    g = torch.randn_like(out)
    out.backward(g)
    gx = x.grad.detach().clone()

    ref = F.conv2d(x.detach().float(), w.detach().float(),
                   b.detach().float(), stride=S)
    ref_relu = F.relu(ref)
    xf = x.detach().float().requires_grad_(True)
    out2 = F.relu(F.conv2d(xf, w.detach().float(), b.detach().float(),
                           stride=S))
    out2.backward(g.float())
    assert torch.allclose(gx.float(), xf.grad, atol=5e-2, rtol=5e-2), \
        (gx.float() - xf.grad).abs().max()


def test_conv_chw_out_matches_flatten(ext):
    """CHW-out epilogue (flatten fused) must equal conv_fwd + torch.flatten,
    forward and backward (mask+transpose kernel)."""
    from distributed_rl_amd.ops import fused_conv_relu

    torch.manual_seed(9)
    N, C, H, W, COUT, KH, S = 16, 64, 9, 9, 64, 3, 1
    x = (torch.randn(N, C, H, W, device=DEV) * 0.5).to(torch.bfloat16
        ).contiguous(memory_format=torch.channels_last).requires_grad_(True)
    x2 = x.detach().clone().requires_grad_(True)
    w = ((torch.randn(COUT, C, KH, KH, device=DEV) * 0.05).to(torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_(True))
    w2 = w.detach().clone().requires_grad_(True)
    b = torch.randn(COUT, device=DEV).to(torch.bfloat16).requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)

    y_chw = fused_conv_relu(x, w, b, S, chw_out=True)
    assert y_chw.dim() == 2
    y_ref = torch.flatten(fused_conv_relu(x2, w2, b2, S), 1)
    assert torch.equal(y_chw, y_ref)

    g = torch.randn_like(y_ref)
    y_chw.backward(g)
    y_ref.backward(g)
    torch.cuda.synchronize()
    assert torch.allclose(x.grad.float(), x2.grad.float(), atol=1e-2), \
        (x.grad.float() - x2.grad.float()).abs().max()
    assert torch.allclose(w.grad.float(), w2.grad.float(), atol=1e-2,
                          rtol=1e-2)
    assert torch.allclose(b.grad.float(), b2.grad.float(), atol=1e-2)
