"""Fused NHWC BatchNorm numerics vs torch.nn.BatchNorm2d fp32 reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs an AMD GPU")


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(8, 64, 16, 16), (4, 96, 7, 5),
                                   (2, 8, 32, 32), (3, 130, 9, 9)])
def test_bn_forward_backward_matches_torch(dtype, shape):
    from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(0)
    N, C, H, W = shape
    dev = torch.device("cuda", 0)

    ref_bn = torch.nn.BatchNorm2d(C).to(dev)
    our_bn = FusedBatchNorm2d.from_batchnorm(ref_bn).to(dev)

    x = torch.randn(N, C, H, W, device=dev)
    x_ref = x.clone().requires_grad_(True)
    x_our = x.clone().to(dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)

    y_ref = ref_bn(x_ref)
    y_our = our_bn(x_our)
    tol = 5e-2 if dtype == torch.bfloat16 else 2e-5
    assert torch.allclose(y_our.float(), y_ref, atol=tol, rtol=tol), \
        (y_our.float() - y_ref).abs().max()

    g = torch.randn_like(y_ref)
    y_ref.backward(g)
    y_our.backward(g.to(dtype).contiguous(memory_format=torch.channels_last))
    torch.cuda.synchronize()

    assert torch.allclose(x_our.grad.float(), x_ref.grad, atol=10 * tol,
                          rtol=10 * tol), \
        (x_our.grad.float() - x_ref.grad).abs().max()
    assert torch.allclose(our_bn.weight.grad, ref_bn.weight.grad,
                          atol=10 * tol, rtol=10 * tol)
    assert torch.allclose(our_bn.bias.grad, ref_bn.bias.grad, atol=10 * tol,
                          rtol=10 * tol)
    # running stats updated identically
    assert torch.allclose(our_bn.running_mean, ref_bn.running_mean, atol=tol,
                          rtol=tol)
    assert torch.allclose(our_bn.running_var, ref_bn.running_var, atol=tol,
                          rtol=tol)


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("with_res", [False, True])
def test_bn_fused_relu_residual(dtype, with_res):
    """y = relu(bn(x) [+ res]) fused in the BN epilogue.

    Forward: vs the fp32 analytic formula. Backward: the correct spec is
    "consistent with our forward's relu mask" — near the kink, bf16
    rounding legitimately picks a different mask than an fp32 reference
    would (the grad differs there by the full dy), so the reference grads
    are computed analytically in fp32 USING our forward's mask."""
    from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(3)
    N, C, H, W = 4, 64, 14, 14
    dev = torch.device("cuda", 0)
    ref_bn = torch.nn.BatchNorm2d(C).to(dev)
    our_bn = FusedBatchNorm2d.from_batchnorm(ref_bn).to(dev)
    eps = ref_bn.eps

    x = torch.randn(N, C, H, W, device=dev)
    res = torch.randn(N, C, H, W, device=dev)
    x_our = x.clone().to(dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    r_our = res.clone().to(dtype).contiguous(
        memory_format=torch.channels_last).requires_grad_(True)

    y_our = our_bn(x_our, relu=True, res=r_our if with_res else None)
    torch.cuda.synchronize()

    # fp32 analytic forward
    m = x.mean((0, 2, 3), keepdim=True)
    v = x.var((0, 2, 3), unbiased=False, keepdim=True)
    istd = torch.rsqrt(v + eps)
    xhat = (x - m) * istd
    w1 = ref_bn.weight.detach().view(1, -1, 1, 1)
    pre = xhat * w1 + ref_bn.bias.detach().view(1, -1, 1, 1)
    if with_res:
        pre = pre + res
    y_ref = torch.relu(pre)
    tol = 5e-2 if dtype == torch.bfloat16 else 2e-5
    assert torch.allclose(y_our.float(), y_ref, atol=tol, rtol=tol), \
        (y_our.float() - y_ref).abs().max()

    g = torch.randn(N, C, H, W, device=dev)
    y_our.backward(g.to(dtype).contiguous(memory_format=torch.channels_last))
    torch.cuda.synchronize()

    # analytic backward with OUR mask
    mask = (y_our.detach().float() > 0)
    gm = g * mask
    M = N * H * W
    sdy = gm.sum((0, 2, 3), keepdim=True)
    sdyx = (gm * xhat).sum((0, 2, 3), keepdim=True)
    dx_ref = w1 * istd * (gm - sdy / M - xhat * sdyx / M)
    dw_ref = sdyx.flatten()
    db_ref = sdy.flatten()

    assert torch.allclose(x_our.grad.float(), dx_ref, atol=10 * tol,
                          rtol=10 * tol), \
        (x_our.grad.float() - dx_ref).abs().max()
    if with_res:
        assert torch.allclose(r_our.grad.float(), gm, atol=10 * tol,
                              rtol=10 * tol), \
            (r_our.grad.float() - gm).abs().max()
    assert torch.allclose(our_bn.weight.grad, dw_ref, atol=10 * tol,
                          rtol=10 * tol), \
        (our_bn.weight.grad - dw_ref).abs().max()
    assert torch.allclose(our_bn.bias.grad, db_ref, atol=10 * tol,
                          rtol=10 * tol)


@requires_gpu
def test_bn_fused_relu_odd_channels():
    """Fused relu on the scalar-fallback path (C not divisible by VEC)."""
    from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(4)
    N, C, H, W = 3, 67, 9, 9
    dev = torch.device("cuda", 0)
    ref_bn = torch.nn.BatchNorm2d(C).to(dev)
    our_bn = FusedBatchNorm2d.from_batchnorm(ref_bn).to(dev)
    x = torch.randn(N, C, H, W, device=dev)
    x_ref = x.clone().requires_grad_(True)
    x_our = x.clone().contiguous(
        memory_format=torch.channels_last).requires_grad_(True)
    y_ref = torch.relu(ref_bn(x_ref))
    y_our = our_bn(x_our, relu=True)
    torch.cuda.synchronize()
    assert torch.allclose(y_our.float(), y_ref, atol=2e-5, rtol=2e-5)
    g = torch.randn_like(y_ref)
    y_ref.backward(g)
    y_our.backward(g.contiguous(memory_format=torch.channels_last))
    torch.cuda.synchronize()
    assert torch.allclose(x_our.grad, x_ref.grad, atol=2e-4, rtol=2e-4)


@requires_gpu
def test_bn_eval_mode():
    from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d
    torch.manual_seed(1)
    dev = torch.device("cuda", 0)
    ref_bn = torch.nn.BatchNorm2d(32).to(dev)
    # give running stats some history
    for _ in range(3):
        ref_bn(torch.randn(4, 32, 8, 8, device=dev))
    our_bn = FusedBatchNorm2d.from_batchnorm(ref_bn).to(dev)
    ref_bn.eval()
    our_bn.eval()
    x = torch.randn(4, 32, 8, 8, device=dev)
    with torch.no_grad():
        y_ref = ref_bn(x)
        y_our = our_bn(x.contiguous(memory_format=torch.channels_last))
    torch.cuda.synchronize()
    assert torch.allclose(y_our, y_ref, atol=1e-4, rtol=1e-4)


@requires_gpu
def test_bn_convert_resnet():
    from torchmpi_amd.models import resnet50
    from torchmpi_amd.ops.batchnorm import FusedBatchNorm2d, convert_batchnorm
    model = resnet50(num_classes=10)
    convert_batchnorm(model)
    n_fused = sum(1 for m in model.modules()
                  if isinstance(m, FusedBatchNorm2d))
    assert n_fused == 53  # all ResNet-50 BN layers replaced
    dev = torch.device("cuda", 0)
    model = model.to(dev).to(memory_format=torch.channels_last)
    x = torch.randn(2, 3, 64, 64, device=dev).contiguous(
        memory_format=torch.channels_last)
    with torch.autocast("cuda", dtype=torch.bfloat16):
        y = model(x)
        loss = y.float().sum()
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)
