"""GPU numerics tests for the fp8 (OCP e4m3, MX-scaled MFMA) path.

The fp8 kernels quantize per tensor (scale folded into the temperature) and
run v_mfma_scale_f32_16x16x128_f8f6f4 with unit block scales.  Reference is
plain PyTorch fp32 on the *quantized-dequantized* values, so tolerances
measure kernel correctness, not quantization error; a separate test bounds
the end-to-end quantization error against the bf16 path.
"""

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


def make_inputs(b, n, d, seed=0):
    g = torch.Generator().manual_seed(seed)
    zi = F.normalize(torch.randn(b, d, generator=g), dim=-1)
    zt = F.normalize(torch.randn(n, d, generator=g), dim=-1)
    tp = torch.tensor(math.log(10.0), device="cuda")
    bs = torch.tensor(-10.0, device="cuda")
    return (zi.cuda().bfloat16(), zt.cuda().bfloat16(), tp, bs)


def dequant_ref(x):
    q, s = ops._quant_fp8(x)
    return q.float() * s


@pytest.mark.parametrize("b,n,diag", [
    (256, 256, 0),
    (512, 1024, 512),
    (300, 272, None),      # ragged edges → guarded fp8 path
])
@pytest.mark.parametrize("d", [128, 768])
def test_fp8_fwd_matches_fp32_on_quantized(b, n, diag, d):
    zi, zt, tp, bs = make_inputs(b, n, d, seed=b + d)
    got = ops.siglip_fwd(zi, zt, tp, bs, diag, quant="fp8")
    want = _torch_loss(dequant_ref(zi), dequant_ref(zt), tp.float(),
                       bs.float(), diag, col_chunk=None)
    torch.cuda.synchronize()
    assert torch.allclose(got, want, rtol=2e-2, atol=1e-2), \
        f"fp8 fwd {got.item()} vs ref-on-quantized {want.item()}"


def test_fp8_fwd_close_to_bf16():
    """End-to-end quantization error stays small on normalized embeddings."""
    zi, zt, tp, bs = make_inputs(1024, 1024, 768, seed=3)
    l8 = ops.siglip_fwd(zi, zt, tp, bs, 0, quant="fp8")
    l16 = ops.siglip_fwd(zi, zt, tp, bs, 0, quant="bf16")
    torch.cuda.synchronize()
    assert torch.allclose(l8, l16, rtol=5e-2), (l8.item(), l16.item())


@pytest.mark.parametrize("b,n,diag", [(256, 512, 0), (512, 512, None)])
def test_fp8_bwd_matches_fp32_on_quantized(b, n, diag):
    d = 256
    zi, zt, tp, bs = make_inputs(b, n, d, seed=9)
    go = torch.tensor(0.61, device="cuda")
    dzi, dzt, dtp, dbs = ops.siglip_bwd(zi, zt, tp, bs, diag, go, None,
                                        quant="fp8")
    # Reference: g from quantized logits, but gradient GEMMs against the
    # original (bf16) embeddings — mirror that with fp32 math.
    zi_q, zt_q = dequant_ref(zi), dequant_ref(zt)
    r_dzi_q, r_dzt_q, r_dtp, r_dbs = _torch_bwd(
        zi_q, zt_q, tp.float(), bs.float(), diag, go.float(), col_chunk=None)
    torch.cuda.synchronize()
    # dzimg/dztxt: kernel uses original bf16 in the GEMMs, reference uses
    # quantized — allow combined quantization tolerance.
    assert torch.allclose(dzi.float(), r_dzi_q, rtol=2e-1, atol=2e-3), \
        (dzi.float() - r_dzi_q).abs().max()
    assert torch.allclose(dzt.float(), r_dzt_q, rtol=2e-1, atol=2e-3)
    assert torch.allclose(dtp.float(), r_dtp, rtol=5e-2, atol=1e-3)
    assert torch.allclose(dbs.float(), r_dbs, rtol=5e-2, atol=1e-3)


def test_fp8_autograd_end_to_end():
    b, n, d = 512, 512, 768
    zi, zt, tp, bs = make_inputs(b, n, d, seed=17)
    zi = zi.clone().requires_grad_(True)
    zt = zt.clone().requires_grad_(True)
    tp = tp.clone().requires_grad_(True)
    bs = bs.clone().requires_grad_(True)
    loss = sigmoid_contrastive_loss(zi, zt, tp, bs, diag_offset=0,
                                    quant="fp8")
    loss.backward()
    torch.cuda.synchronize()
    for grad in (zi.grad, zt.grad, tp.grad, bs.grad):
        assert grad is not None
        assert torch.isfinite(grad.float()).all()

    # Grad direction must agree with the bf16 path.
    zi2 = zi.detach().clone().requires_grad_(True)
    zt2 = zt.detach().clone().requires_grad_(True)
    loss2 = sigmoid_contrastive_loss(zi2, zt2, tp.detach(), bs.detach(),
                                     diag_offset=0, quant="bf16")
    loss2.backward()
    torch.cuda.synchronize()
    cos = F.cosine_similarity(zi.grad.float().flatten(),
                              zi2.grad.float().flatten(), dim=0)
    assert cos > 0.98, cos.item()


def test_fp8_d_validation():
    zi = torch.randn(32, 24, device="cuda").bfloat16()  # d%16 != 0
    tp = torch.tensor(0.0, device="cuda")
    with pytest.raises(RuntimeError, match="multiple of 16"):
        ops.siglip_fwd(zi, zi, tp, tp, 0, quant="fp8")


def test_fp8_chunked_matches_single_slab():
    b, n, d = 256, 1024, 256
    zi, zt, tp, bs = make_inputs(b, n, d, seed=31)
    go = torch.tensor(1.0, device="cuda")
    full = ops.siglip_bwd(zi, zt, tp, bs, 128, go, None, quant="fp8")
    chunked = ops.siglip_bwd(zi, zt, tp, bs, 128, go, 256, quant="fp8")
    torch.cuda.synchronize()
    for a, b_ in zip(full, chunked):
        assert torch.allclose(a.float(), b_.float(), rtol=3e-2, atol=1e-3)


def test_fp8_bwd_fallback_unaligned_shapes():
    """b%16 != 0 disables the _scaled_mm path → dequantized-g rocBLAS
    fallback (and the guarded kernel path for the ragged tile)."""
    b, n, d = 260, 512, 256
    zi, zt, tp, bs = make_inputs(b, n, d, seed=41)
    go = torch.tensor(1.0, device="cuda")
    dzi, dzt, dtp, dbs = ops.siglip_bwd(zi, zt, tp, bs, 100, go, None,
                                        quant="fp8")
    zi_q, zt_q = dequant_ref(zi), dequant_ref(zt)
    r = _torch_bwd(zi_q, zt_q, tp.float(), bs.float(), 100, go.float(),
                   col_chunk=None)
    torch.cuda.synchronize()

    def rel_l2(a, b_):
        return ((a.float() - b_).norm() / b_.norm().clamp(min=1e-12)).item()

    assert rel_l2(dzi, r[0]) < 5e-2
    assert rel_l2(dzt, r[1]) < 5e-2
    assert torch.allclose(dtp.float(), r[2], rtol=5e-2, atol=2e-3)
    assert torch.allclose(dbs.float(), r[3], rtol=5e-2, atol=2e-3)


def test_mixed_policy_bwd_matches_bf16():
    """quant='mixed': bf16 logits with fp8 gradient GEMMs — grads must stay
    close to the pure-bf16 path (g and GEMM operands quantized e4m3)."""
    b, n, d = 512, 512, 768
    zi, zt, tp, bs = make_inputs(b, n, d, seed=55)
    go = torch.tensor(1.0, device="cuda")
    mixed = ops.siglip_bwd(zi, zt, tp, bs, 0, go, None, quant="mixed")
    ref = ops.siglip_bwd(zi, zt, tp, bs, 0, go, None, quant="bf16")
    l_mixed = ops.siglip_fwd(zi, zt, tp, bs, 0, quant="mixed")
    l_bf16 = ops.siglip_fwd(zi, zt, tp, bs, 0, quant="bf16")
    torch.cuda.synchronize()
    # identical forward (mixed logits ARE bf16)
    assert torch.allclose(l_mixed, l_bf16, rtol=1e-5, atol=1e-3)

    def rel_l2(a, b_):
        bf = b_.float()
        return ((a.float() - bf).norm() / bf.norm().clamp(min=1e-12)).item()

    assert rel_l2(mixed[0], ref[0]) < 3e-2
    assert rel_l2(mixed[1], ref[1]) < 3e-2
    assert torch.allclose(mixed[2].float(), ref[2].float(), rtol=2e-2,
                          atol=1e-3)
    assert torch.allclose(mixed[3].float(), ref[3].float(), rtol=2e-2,
                          atol=1e-3)
