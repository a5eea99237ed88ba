"""HIP kernel dispatch for the fused SigLIP loss (MI355X / gfx950).

Loads the in-tree ``_siglip_hip.so`` (built by ``ops.build``) via ctypes and
exposes:

- :func:`siglip_fwd` — fused logits+logsigmoid+sum forward over a (b, n)
  block; the logits matrix never leaves MFMA accumulators.
- :func:`siglip_bwd` — backward: fused recompute kernel emits the g slab and
  the scalar partials; the two ``(b,n)×(n,d)`` gradient GEMMs run on rocBLAS
  via ``torch.matmul`` (plain library GEMMs), column-chunked so workspace is
  O(b · col_chunk) regardless of n.

These are *loud* paths: calling them on a GPU without the built extension
raises — there is no silent eager fallback on device (CPU fallbacks live in
``losses/functional.py`` and are CPU-only by construction).
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

from .build import SO_PATH, build as build_extension

_DIAG_NONE = -(2 ** 31)


def _kernel_flags() -> int:
    # bit 0: XCD-contiguous block remap (default off — measured slightly
    #        negative in the L3-resident regime; SIGLIP_XCD_SWZ=1 enables).
    # bit 1: grouped block walk for L2 panel reuse (default on;
    #        SIGLIP_GROUP_SWZ=0 disables for A/B profiling).
    f = 0
    if os.environ.get("SIGLIP_XCD_SWZ", "1") != "0":
        f |= 1
    if os.environ.get("SIGLIP_GROUP_SWZ", "1") != "0":
        f |= 2
    return f

_lib = None
_lib_err: Optional[str] = None


def _load():
    global _lib, _lib_err
    if _lib is not None or _lib_err is not None:
        return _lib
    if not os.path.exists(SO_PATH):
        _lib_err = (
            f"HIP extension not found at {SO_PATH}. Build it with: "
            "python -m distributed_sigmoid_loss_amd.ops.build")
        return None
    try:
        lib = ctypes.CDLL(SO_PATH)
    except OSError as e:  # pragma: no cover
        _lib_err = f"failed to load {SO_PATH}: {e}"
        return None
    lib.siglip_ext_abi.restype = ctypes.c_int
    lib.siglip_fwd_bf16.restype = ctypes.c_int
    lib.siglip_fwd_bf16.argtypes = [ctypes.c_void_p] * 6 + [ctypes.c_int] * 5
    lib.siglip_bwd_g_bf16.restype = ctypes.c_int
    lib.siglip_bwd_g_bf16.argtypes = [ctypes.c_void_p] * 7 + [ctypes.c_int] * 5
    _lib = lib
    return _lib


def extension_available() -> bool:
    return _load() is not None


def _require_lib():
    lib = _load()
    if lib is None:
        raise RuntimeError(_lib_err)
    return lib


def _check(rc: int, what: str):
    if rc != 0:
        raise RuntimeError(f"{what} failed: hipError_t={rc}")


def _prep_scalar(p: torch.Tensor, device) -> torch.Tensor:
    t = p.detach().reshape(()).to(device=device, dtype=torch.float32)
    return t.contiguous()


def _validate(zimg: torch.Tensor, ztxt: torch.Tensor):
    if not zimg.is_cuda:
        raise RuntimeError("siglip HIP ops require GPU tensors")
    if zimg.dtype != torch.bfloat16 or ztxt.dtype != torch.bfloat16:
        raise RuntimeError(
            f"siglip HIP ops require bf16 embeddings (got {zimg.dtype}); "
            "cast with .bfloat16() or pass impl='torch'")
    if zimg.shape[1] % 8 != 0:
        raise RuntimeError(f"emb dim must be a multiple of 8 (got {zimg.shape[1]})")


def siglip_fwd(zimg: torch.Tensor, ztxt: torch.Tensor, t_prime: torch.Tensor,
               bias: torch.Tensor, diag_offset: Optional[int]) -> torch.Tensor:
    lib = _require_lib()
    _validate(zimg, ztxt)
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    bp = _prep_scalar(bias, dev)
    loss = torch.zeros((), device=dev, dtype=torch.float32)
    stream = torch.cuda.current_stream(dev).cuda_stream
    diag = _DIAG_NONE if diag_offset is None else int(diag_offset)
    _check(lib.siglip_fwd_bf16(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(zimg.data_ptr()), ctypes.c_void_p(ztxt.data_ptr()),
        ctypes.c_void_p(tp.data_ptr()), ctypes.c_void_p(bp.data_ptr()),
        ctypes.c_void_p(loss.data_ptr()), b, n, d, diag, _kernel_flags()),
        "siglip_fwd_bf16")
    return loss


def siglip_bwd(zimg: torch.Tensor, ztxt: torch.Tensor, t_prime: torch.Tensor,
               bias: torch.Tensor, diag_offset: Optional[int],
               grad_output: torch.Tensor, col_chunk: int):
    """Returns (dzimg, dztxt, dt_prime, dbias).

    Per column slab: the fused kernel recomputes logit tiles (MFMA) and writes
    g = dL/d(logit pre-scale); then dzimg += g @ ztxt_slab and
    dztxt_slab = gᵀ @ zimg (rocBLAS bf16 GEMMs, fp32 accumulation buffers).
    """
    lib = _require_lib()
    _validate(zimg, ztxt)
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    bp = _prep_scalar(bias, dev)
    scal = torch.zeros(2, device=dev, dtype=torch.float32)
    stream = torch.cuda.current_stream(dev).cuda_stream

    # Column slab sizing: the kernel's g-store addressing is 32-bit, so
    # b * slab * 2 bytes must stay below 2^32; beyond that (and to bound
    # workspace at huge n) we chunk.
    step = col_chunk if col_chunk and col_chunk > 0 else n
    while (b * step * 2) >= 2 ** 32:
        step //= 2
    step = max(step, 256)

    t = tp.exp()
    go = grad_output.detach().reshape(()).to(device=dev, dtype=torch.float32)
    scale = go * t

    def run_g(zt_slab, g_slab, diag):
        _check(lib.siglip_bwd_g_bf16(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(zimg.data_ptr()),
            ctypes.c_void_p(zt_slab.data_ptr()),
            ctypes.c_void_p(tp.data_ptr()), ctypes.c_void_p(bp.data_ptr()),
            ctypes.c_void_p(g_slab.data_ptr()),
            ctypes.c_void_p(scal.data_ptr()),
            b, zt_slab.shape[0], d, diag, _kernel_flags()),
            "siglip_bwd_g_bf16")

    if step >= n:
        # Single-slab fast path: no fp32 accumulation round trips.
        g = torch.empty((b, n), device=dev, dtype=torch.bfloat16)
        run_g(ztxt, g, _DIAG_NONE if diag_offset is None else int(diag_offset))
        dzimg = (g @ ztxt) * scale
        dztxt = (g.T @ zimg) * scale
    else:
        dzimg_acc = torch.zeros((b, d), device=dev, dtype=torch.float32)
        dztxt = torch.empty((n, d), device=dev, dtype=ztxt.dtype)
        g_buf = torch.empty((b, step), device=dev, dtype=torch.bfloat16)
        for j0 in range(0, n, step):
            j1 = min(j0 + step, n)
            c = j1 - j0
            zt = ztxt[j0:j1]
            g = g_buf if c == step else torch.empty(
                (b, c), device=dev, dtype=torch.bfloat16)
            diag = _DIAG_NONE if diag_offset is None else int(diag_offset) - j0
            run_g(zt, g, diag)
            dzimg_acc += (g @ zt).float()
            dztxt[j0:j1] = (g.T @ zimg) * scale
        dzimg = dzimg_acc * scale

    dzimg = dzimg.to(zimg.dtype)
    dztxt = dztxt.to(ztxt.dtype)
    dt_prime = (scal[0] * scale).to(t_prime.dtype).reshape(t_prime.shape)
    dbias = (scal[1] * go).to(bias.dtype).reshape(bias.shape)
    return dzimg, dztxt, dt_prime, dbias
