"""GPU convergence: optimizing through the fused kernels reduces the loss
(the reference's qualitative claim, exercised end to end per policy)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("needs a GPU", allow_module_level=True)

from distributed_sigmoid_loss_amd import DistributedSigmoidLoss
from distributed_sigmoid_loss_amd.models import TwoTowerModel


@pytest.mark.parametrize("quant", ["bf16", "fp8", "mixed"])
def test_loss_decreases(quant):
    torch.manual_seed(7)
    b, d = 2048, 256
    model = TwoTowerModel(d, d).to(device="cuda", dtype=torch.bfloat16)
    loss_mod = DistributedSigmoidLoss(b, quant=quant).cuda()
    opt = torch.optim.AdamW(
        list(model.parameters()) + list(loss_mod.parameters()), lr=2e-3)
    img = torch.randn(b, d, device="cuda", dtype=torch.bfloat16)
    txt = img + 0.1 * torch.randn_like(img)   # learnable correspondence

    losses = []
    for step in range(60):
        opt.zero_grad(set_to_none=True)
        zi, zt = model(img, txt)
        loss = loss_mod(zi, zt)
        loss.backward()
        opt.step()
        if step % 10 == 0 or step == 59:
            losses.append(float(loss.detach()))
    torch.cuda.synchronize()
    assert all(v == v for v in losses), losses          # finite
    assert losses[-1] < losses[0] * 0.7, losses         # ≥30% reduction
