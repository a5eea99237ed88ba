import sys, os, ctypes
sys.path.insert(0, "/root/repo")
import torch, torch.nn.functional as F
from distributed_sigmoid_loss_amd import ops

def time_fn(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True); e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters

b, d = 32768, 768
zi = F.normalize(torch.randn(b, d, device="cuda"), dim=-1).bfloat16()
lib = ops._require_lib()
q = torch.empty((b, d), device="cuda", dtype=torch.float8_e4m3fn)
e8 = torch.empty(b, device="cuda", dtype=torch.uint8)
emax = torch.zeros(1, device="cuda", dtype=torch.int32)
stream = torch.cuda.current_stream().cuda_stream
def raw():
    lib.quant_fp8_rowwise_bf16(ctypes.c_void_p(stream),
        ctypes.c_void_p(zi.data_ptr()), ctypes.c_void_p(q.data_ptr()),
        ctypes.c_void_p(e8.data_ptr()), ctypes.c_void_p(emax.data_ptr()), b, d)
print(f"raw rowwise kernel    : {time_fn(raw):7.3f} ms")
def post():
    emax_f = emax[0].float()
    ratio = torch.exp2(e8.float() - emax_f)
    s_ref = torch.exp2(emax_f - 127.0)
    return ratio, s_ref
print(f"torch post-ops        : {time_fn(post):7.3f} ms")
print(f"full _quant_fp8_rowwise: {time_fn(lambda: ops._quant_fp8_rowwise(zi)):6.3f} ms")
print(f"per-tensor _quant_fp8 : {time_fn(lambda: ops._quant_fp8(zi)):7.3f} ms")
