import math, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.nn.functional as F
from distributed_sigmoid_loss_amd import ops
from distributed_sigmoid_loss_amd.losses.functional import _torch_loss, _torch_bwd

for case in [(452, 120, 8, -360), (243, 18, 8, 0), (36, 86, 16, 0)]:
    b, n, d, diag = case
    g = torch.Generator().manual_seed(hash(case) & 0xFFFF)
    zi = F.normalize(torch.randn(b, d, generator=g), dim=-1).cuda().bfloat16()
    zt = F.normalize(torch.randn(n, d, generator=g), dim=-1).cuda().bfloat16()
    tp = torch.tensor(math.log(7.0), device="cuda")
    bs = torch.tensor(-6.0, device="cuda")
    go = torch.tensor(0.9, device="cuda")
    got = ops.siglip_fwd(zi, zt, tp, bs, diag)
    want = _torch_loss(zi.float(), zt.float(), tp.float(), bs.float(), diag, None)
    dk = ops.siglip_bwd(zi, zt, tp, bs, diag, go, None)
    dr = _torch_bwd(zi.float(), zt.float(), tp.float(), bs.float(), diag, go.float(), None)
    torch.cuda.synchronize()
    print(f"case {case}: fwd got {got.item():.6f} want {want.item():.6f} relerr {abs(got.item()-want.item())/abs(want.item()):.2e}")
    for name, a, r in zip(["dzi","dzt","dtp","dbs"], dk, dr):
        a = a.float(); r = r.float()
        bad = (~torch.isclose(a, r, rtol=6e-2, atol=2e-3)).sum().item()
        print(f"  {name}: maxabs diff {(a-r).abs().max().item():.4e}  ref maxabs {r.abs().max().item():.4e}  n_bad {bad}/{r.numel()}")
        if bad and r.numel() > 1:
            idx = (a-r).abs().argmax().item()
            print(f"    worst at flat {idx}: got {a.flatten()[idx].item():.5f} ref {r.flatten()[idx].item():.5f}")
