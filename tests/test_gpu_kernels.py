"""GPU numerics tests: hand-written CDNA4 kernels vs plain PyTorch fp32
reference of the same op.  Run with ``pytest -m gpu`` on an MI355X box."""

import math

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("needs a GPU", allow_module_level=True)

from distributed_sigmoid_loss_amd import ops
from distributed_sigmoid_loss_amd.losses.functional import (
    sigmoid_contrastive_loss,
    _torch_loss,
    _torch_bwd,
)


def test_extension_loaded():
    """The native path must be present on a GPU box — no silent fallback."""
    assert ops.extension_available(), ops._lib_err


def make_inputs(b, n, d, seed=0, device="cuda"):
    g = torch.Generator().manual_seed(seed)
    zi = F.normalize(torch.randn(b, d, generator=g), dim=-1)
    zt = F.normalize(torch.randn(n, d, generator=g), dim=-1)
    tp = torch.tensor(math.log(10.0))
    bs = torch.tensor(-10.0)
    return (zi.to(device=device, dtype=torch.bfloat16),
            zt.to(device=device, dtype=torch.bfloat16),
            tp.to(device), bs.to(device))


def fp32_ref_loss(zi_bf16, zt_bf16, tp, bs, diag):
    """fp32 CPU-math reference on the same (bf16-rounded) values."""
    return _torch_loss(zi_bf16.float(), zt_bf16.float(), tp.float(),
                       bs.float(), diag, col_chunk=None)


@pytest.mark.parametrize("b,n,diag", [
    (128, 128, 0),
    (256, 512, 256),        # own-chunk offset inside a wider block
    (300, 260, None),       # ragged edges, negatives-only
    (1024, 2048, 1024),
    (73, 129, -5),          # sub-tile sizes + negative offset
])
@pytest.mark.parametrize("d", [64, 768])
def test_fwd_kernel_matches_fp32(b, n, diag, d):
    zi, zt, tp, bs = make_inputs(b, n, d, seed=b + d)
    got = ops.siglip_fwd(zi, zt, tp, bs, diag)
    want = fp32_ref_loss(zi, zt, tp, bs, diag).to(got.device)
    torch.cuda.synchronize()
    # bf16 MFMA with fp32 accumulate vs fp32 math: loose elementwise rounding
    # but the sum is large and errors average out.
    assert torch.allclose(got, want, rtol=2e-2, atol=1e-2), \
        f"fwd loss {got.item()} vs ref {want.item()}"


@pytest.mark.parametrize("b,n,diag", [
    (128, 128, 0),
    (300, 260, None),
    (512, 1024, 512),
])
@pytest.mark.parametrize("col_chunk", [None, 192])
def test_bwd_kernel_matches_fp32(b, n, diag, col_chunk):
    d = 256
    zi, zt, tp, bs = make_inputs(b, n, d, seed=7)
    go = torch.tensor(0.73, device="cuda")
    dzi, dzt, dtp, dbs = ops.siglip_bwd(zi, zt, tp, bs, diag, go,
                                        col_chunk or n)
    r_dzi, r_dzt, r_dtp, r_dbs = _torch_bwd(
        zi.float(), zt.float(), tp.float(), bs.float(), diag, go.float(),
        col_chunk=None)
    torch.cuda.synchronize()
    assert torch.allclose(dzi.float(), r_dzi, rtol=5e-2, atol=5e-4), \
        (dzi.float() - r_dzi).abs().max()
    assert torch.allclose(dzt.float(), r_dzt, rtol=5e-2, atol=5e-4), \
        (dzt.float() - r_dzt).abs().max()
    assert torch.allclose(dtp.float(), r_dtp, rtol=2e-2, atol=1e-3)
    assert torch.allclose(dbs.float(), r_dbs, rtol=2e-2, atol=1e-3)


def test_autograd_function_end_to_end():
    """Full autograd through the fused op on GPU vs fp32 reference."""
    b, n, d = 384, 384, 768
    zi, zt, tp, bs = make_inputs(b, n, d, seed=11)
    zi = zi.clone().requires_grad_(True)
    zt = zt.clone().requires_grad_(True)
    tp = tp.clone().requires_grad_(True)
    bs = bs.clone().requires_grad_(True)
    loss = sigmoid_contrastive_loss(zi, zt, tp, bs, diag_offset=0)
    loss.backward()

    zi2 = zi.detach().float().clone().requires_grad_(True)
    zt2 = zt.detach().float().clone().requires_grad_(True)
    tp2 = tp.detach().float().clone().requires_grad_(True)
    bs2 = bs.detach().float().clone().requires_grad_(True)
    ref = _torch_loss(zi2, zt2, tp2, bs2, 0, col_chunk=None)
    ref.backward()
    torch.cuda.synchronize()

    assert torch.allclose(loss, ref, rtol=2e-2, atol=1e-2)
    assert torch.allclose(zi.grad.float(), zi2.grad, rtol=5e-2, atol=5e-4)
    assert torch.allclose(zt.grad.float(), zt2.grad, rtol=5e-2, atol=5e-4)
    assert torch.allclose(tp.grad.float(), tp2.grad, rtol=2e-2, atol=1e-3)
    assert torch.allclose(bs.grad.float(), bs2.grad, rtol=2e-2, atol=1e-3)


def test_fp32_input_raises():
    """Loud failure on non-bf16 GPU inputs — no silent eager fallback."""
    zi = torch.randn(16, 64, device="cuda")
    zt = torch.randn(16, 64, device="cuda")
    tp = torch.tensor(0.0, device="cuda")
    with pytest.raises(RuntimeError, match="bf16"):
        ops.siglip_fwd(zi, zt, tp, tp, 0)


def test_single_gpu_module_matches_cpu():
    """DistributedSigmoidLoss on one GPU (no process group) vs CPU fp32."""
    from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
    b, d = 256, 768
    g = torch.Generator().manual_seed(5)
    zi32 = F.normalize(torch.randn(b, d, generator=g), dim=-1)
    zt32 = F.normalize(torch.randn(b, d, generator=g), dim=-1)

    mod_gpu = DistributedSigmoidLoss(b).cuda()
    zi = zi32.cuda().bfloat16().requires_grad_(True)
    zt = zt32.cuda().bfloat16().requires_grad_(True)
    loss = mod_gpu(zi, zt)
    loss.backward()

    mod_cpu = DistributedSigmoidLoss(b)
    zi_c = zi32.clone().requires_grad_(True)
    zt_c = zt32.clone().requires_grad_(True)
    loss_c = mod_cpu(zi_c, zt_c)
    loss_c.backward()
    torch.cuda.synchronize()

    assert torch.allclose(loss.cpu().float(), loss_c, rtol=2e-2, atol=1e-2)
    assert torch.allclose(zi.grad.cpu().float(), zi_c.grad, rtol=5e-2,
                          atol=5e-4)
    assert torch.allclose(mod_gpu.t_prime.grad.cpu(), mod_cpu.t_prime.grad,
                          rtol=2e-2, atol=1e-3)


@pytest.mark.parametrize("quant", ["bf16"])
def test_chunked_matches_unchunked_gpu(quant):
    """Column-chunked backward (the huge-batch path, BASELINE config 4) is
    numerically consistent with the single-slab path on GPU."""
    b, n, d = 512, 4096, 256
    zi, zt, tp, bs = make_inputs(b, n, d, seed=21)
    go = torch.tensor(1.0, device="cuda")
    full = ops.siglip_bwd(zi, zt, tp, bs, 100, go, None)
    chunked = ops.siglip_bwd(zi, zt, tp, bs, 100, go, 1024)
    torch.cuda.synchronize()
    for a, b_ in zip(full, chunked):
        assert torch.allclose(a.float(), b_.float(), rtol=2e-2, atol=5e-4)


def test_extension_is_intree_so():
    """The loaded native library must be the in-tree .so (driver checks which
    .so files the process actually loaded)."""
    import distributed_sigmoid_loss_amd
    pkg_root = distributed_sigmoid_loss_amd.__file__.rsplit("/", 2)[0]
    assert ops.SO_PATH.startswith(pkg_root)
    ops._require_lib()
    with open("/proc/self/maps") as f:
        assert any("_siglip_hip.so" in line for line in f)


@pytest.mark.parametrize("b,d", [(256, 768), (1000, 66), (8, 2)])
def test_fused_l2_normalize(b, d):
    """Fused single-pass normalize kernels vs F.normalize autograd chain."""
    g = torch.Generator().manual_seed(b)
    x32 = torch.randn(b, d, generator=g) * 3.0
    x = x32.cuda().bfloat16().requires_grad_(True)
    y = ops.l2_normalize(x)
    go = torch.randn(b, d, generator=g).cuda().bfloat16()
    y.backward(go)

    x2 = x.detach().clone().requires_grad_(True)
    y2 = F.normalize(x2.float(), dim=-1)
    y2.backward(go.float())
    torch.cuda.synchronize()

    assert torch.allclose(y.float(), y2.detach(), rtol=1e-2, atol=1e-2)
    assert torch.allclose(x.grad.float(), x2.grad.float(), rtol=5e-2,
                          atol=1e-2), \
        (x.grad.float() - x2.grad.float()).abs().max()


def test_siglip_loss_module_gpu():
    """SigLipLoss (caller-owned params, reference rwightman API) through the
    fused kernels on one GPU vs CPU fp32."""
    from distributed_sigmoid_loss_amd import SigLipLoss
    b, d = 256, 768
    g = torch.Generator().manual_seed(12)
    zi32 = F.normalize(torch.randn(b, d, generator=g), dim=-1)
    zt32 = F.normalize(torch.randn(b, d, generator=g), dim=-1)
    scale32 = torch.tensor(math.log(10.0))
    bias32 = torch.tensor(-10.0)

    mod = SigLipLoss(rank=0, world_size=1)
    zi = zi32.cuda().bfloat16().requires_grad_(True)
    zt = zt32.cuda().bfloat16().requires_grad_(True)
    sc = scale32.cuda().requires_grad_(True)
    bs = bias32.cuda().requires_grad_(True)
    loss = mod(zi, zt, sc, bs)
    loss.backward()

    zi_c = zi32.clone().requires_grad_(True)
    zt_c = zt32.clone().requires_grad_(True)
    sc_c = scale32.clone().requires_grad_(True)
    bs_c = bias32.clone().requires_grad_(True)
    loss_c = mod(zi_c, zt_c, sc_c, bs_c)
    loss_c.backward()
    torch.cuda.synchronize()

    assert torch.allclose(loss.cpu().float(), loss_c, rtol=2e-2, atol=1e-2)
    assert torch.allclose(zi.grad.cpu().float(), zi_c.grad, rtol=5e-2,
                          atol=5e-4)
    assert torch.allclose(sc.grad.cpu().float(), sc_c.grad, rtol=2e-2,
                          atol=1e-3)
    assert torch.allclose(bs.grad.cpu().float(), bs_c.grad, rtol=2e-2,
                          atol=1e-3)
