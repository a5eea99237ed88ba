"""HIP kernel dispatch for the fused SigLIP loss (MI355X / gfx950).

Loads the in-tree ``_siglip_hip.so`` (built by ``ops.build``) via ctypes and
exposes:

- :func:`siglip_fwd` — fused logits+logsigmoid+sum forward over a (b, n)
  block; the logits matrix never leaves MFMA accumulators.
- :func:`siglip_fwd_g` / :func:`siglip_bwd_from_g` — the default training
  pair ("saved-g"): one kernel emits loss + the dL/dlogit slab + both
  scalar partials, and backward is just the two ``(b,n)×(n,d)`` gradient
  GEMMs on the libraries (rocBLAS bf16 via ``torch.matmul``, tuned
  hipBLASLt fp8 via ``torch._scaled_mm``).
- :func:`siglip_bwd` — recompute backward for explicit ``col_chunk`` runs
  and batches whose g slab would not fit: the recompute kernel re-derives
  logit tiles and emits g slab by slab, O(b · col_chunk) workspace.
- quantization policies: ``bf16`` | ``fp8`` (e4m3 logits via the MX-scaled
  MFMA — per-tensor scales folded into the temperature, or per-row e8m0
  hardware dequant with ``SIGLIP_FP8_ROWWISE=1``) | ``mixed`` (bf16
  logits, fp8 gradient GEMMs).
- fused tower helpers: :func:`l2_normalize` (single-pass fwd/bwd) and the
  fused fp8 quantizers.

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

# Kernel scalar outputs land in a zeroed (8, 32)-float buffer — one 128-B
# cache-line slot per XCD (per-block atomics stay XCD-local; a single shared
# line measured as a serialized ~10ns/op drain).  Slot layout:
# [0] loss, [1] Σg·dot, [2] Σg.  Reduce with the helpers below.
_OUT_SLOTS, _OUT_STRIDE = 8, 32


def _out_buf(device) -> torch.Tensor:
    return torch.zeros((_OUT_SLOTS, _OUT_STRIDE), device=device,
                       dtype=torch.float32)


def reduce_out3(buf: torch.Tensor) -> torch.Tensor:
    """Sum the per-XCD slots of a kernel scalar-output buffer →
    ``[loss, Σg·dot, Σg]`` (whichever the mode wrote; others are 0)."""
    return buf[:, :3].sum(dim=0)


def _kernel_flags(default_gm: str = "4") -> int:
    # bit 0: XCD-contiguous block remap — default ON: together with the
    #        grouped walk it measured best (+4%, profiles/flag_matrix_gm4.log
    #        and profiles/README.md); SIGLIP_XCD_SWZ=0 disables.
    # bit 1: grouped block walk for L2 panel reuse (default on;
    #        SIGLIP_GROUP_SWZ=0 disables for A/B profiling).
    # bit 2: non-temporal g/gᵀ slab stores (SIGLIP_NT_G=1; default off —
    #        measured a regression on every mode, profiles round 2).
    # bits 4-5: GROUP_M locality-group height.  Per-mode defaults from the
    #        round-2 sweep (gpurun sweep2): plain fwd is fastest at gm=8
    #        (854 TF), the g-emitting modes at gm=4 (the slab stores change
    #        the L2 picture); SIGLIP_GROUP_M overrides both.
    f = 0
    if os.environ.get("SIGLIP_XCD_SWZ", "1") != "0":
        f |= 1
    if os.environ.get("SIGLIP_GROUP_SWZ", "1") != "0":
        f |= 2
    if os.environ.get("SIGLIP_NT_G", "0") == "1":
        f |= 4
    gm = os.environ.get("SIGLIP_GROUP_M", default_gm)
    f |= {"8": 0, "1": 1, "4": 2, "16": 3}.get(gm, 0) << 4
    return f

_lib = None
_lib_err: Optional[str] = None
_tuned_loaded = False


def load_tuned_gemms() -> bool:
    """Point torch's TunableOp at the shipped MI355X-tuned GEMM table so
    ``torch._scaled_mm`` picks the tuned hipBLASLt kernel for the fp8 grad
    GEMMs (measured 0.52 vs 0.99 ms at the B=32k shape; bf16 rocBLAS
    defaults were already optimal, so this is loaded only for the
    fp8/mixed policies).  Tuning itself stays OFF — shapes missing from
    the table use the normal heuristics."""
    global _tuned_loaded
    if _tuned_loaded:
        return True
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING", "0") == "1":
        # Explicit tuning session (e.g. regenerating the table): leave
        # torch's env-driven TunableOp configuration alone.
        return False
    if not torch.cuda.is_available():
        return False
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                        "tunableop_mi355x.csv")
    if not os.path.exists(path):
        return False
    try:
        t = torch.cuda.tunable
        t.enable(True)
        t.tuning_enable(False)
        t.read_file(path)
        _tuned_loaded = True
    except Exception:  # pragma: no cover - tunable API availability
        return False
    return True


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
    if lib.siglip_ext_abi() != 9:
        _lib_err = (f"stale HIP extension at {SO_PATH} "
                    f"(ABI {lib.siglip_ext_abi()}, need 9); rebuild with: "
                    "python -m distributed_sigmoid_loss_amd.ops.build --force")
        return None
    lib.siglip_fwd_bf16.restype = ctypes.c_int
    lib.siglip_fwd_bf16.argtypes = [ctypes.c_void_p] * 6 + [ctypes.c_int] * 5
    lib.siglip_bwd_g_bf16.restype = ctypes.c_int
    lib.siglip_bwd_g_bf16.argtypes = [ctypes.c_void_p] * 7 + [ctypes.c_int] * 5
    lib.siglip_fwd_fp8.restype = ctypes.c_int
    lib.siglip_fwd_fp8.argtypes = ([ctypes.c_void_p] * 6
                                   + [ctypes.c_int] * 5
                                   + [ctypes.c_void_p] * 2)
    lib.siglip_bwd_g_fp8.restype = ctypes.c_int
    lib.siglip_bwd_g_fp8.argtypes = ([ctypes.c_void_p] * 8
                                     + [ctypes.c_int] * 5
                                     + [ctypes.c_void_p] * 4)
    lib.siglip_bwd_g_mixed.restype = ctypes.c_int
    lib.siglip_bwd_g_mixed.argtypes = (
        [ctypes.c_void_p] * 8 + [ctypes.c_int] * 5)
    lib.siglip_fwdg_bf16.restype = ctypes.c_int
    lib.siglip_fwdg_bf16.argtypes = [ctypes.c_void_p] * 7 + [ctypes.c_int] * 6
    lib.siglip_fwdg_mixed.restype = ctypes.c_int
    lib.siglip_fwdg_mixed.argtypes = (
        [ctypes.c_void_p] * 8 + [ctypes.c_int] * 6)
    lib.siglip_fwdg_fp8.restype = ctypes.c_int
    lib.siglip_fwdg_fp8.argtypes = ([ctypes.c_void_p] * 8
                                    + [ctypes.c_int] * 6
                                    + [ctypes.c_void_p] * 4)
    lib.l2norm_fwd_bf16.restype = ctypes.c_int
    lib.l2norm_fwd_bf16.argtypes = (
        [ctypes.c_void_p] * 4 + [ctypes.c_int] * 2 + [ctypes.c_float])
    lib.l2norm_bwd_bf16.restype = ctypes.c_int
    lib.l2norm_bwd_bf16.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_int] * 2
    lib.quant_fp8_bf16.restype = ctypes.c_int
    lib.quant_fp8_bf16.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_longlong]
    lib.quant_fp8_rowwise_bf16.restype = ctypes.c_int
    lib.quant_fp8_rowwise_bf16.argtypes = (
        [ctypes.c_void_p] * 5 + [ctypes.c_int] * 2)
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


def _validate(zimg: torch.Tensor, ztxt: torch.Tensor, quant: str):
    if not zimg.is_cuda:
        raise RuntimeError("siglip HIP ops require GPU tensors")
    if zimg.dtype != torch.bfloat16 or ztxt.dtype != torch.bfloat16:
        raise RuntimeError(
            f"siglip HIP ops require bf16 embeddings (got {zimg.dtype}); "
            "cast with .bfloat16() or pass impl='torch'")
    if quant not in ("bf16", "fp8", "mixed"):
        raise ValueError(f"unknown quant {quant!r}")
    mult = 16 if quant == "fp8" else 8
    if zimg.shape[1] % mult != 0:
        raise RuntimeError(
            f"emb dim must be a multiple of {mult} for {quant} "
            f"(got {zimg.shape[1]})")


def _quant_fp8(x: torch.Tensor):
    """Per-tensor symmetric quantization to OCP e4m3 (max normal 448).

    Returns (q, scale): x ≈ q * scale.  The scale is folded into the
    temperature for the logit kernels (t_eff = t·s_img·s_txt), so the HIP
    side runs with unit MX block scales.  On GPU the fused two-pass HIP
    kernels run (amax + cast, no host sync — ~0.04 ms vs ~0.25 ms of stock
    kernels per pair at B=32k); elsewhere the torch composite.
    """
    if x.is_cuda and x.dtype == torch.bfloat16 and extension_available():
        lib = _require_lib()
        xd = x.detach().contiguous()
        q = torch.empty(xd.shape, device=xd.device,
                        dtype=torch.float8_e4m3fn)
        scale = torch.empty((), device=xd.device, dtype=torch.float32)
        amax_bits = torch.zeros(1, device=xd.device, dtype=torch.int32)
        stream = torch.cuda.current_stream(xd.device).cuda_stream
        _check(lib.quant_fp8_bf16(
            ctypes.c_void_p(stream), ctypes.c_void_p(xd.data_ptr()),
            ctypes.c_void_p(q.data_ptr()), ctypes.c_void_p(scale.data_ptr()),
            ctypes.c_void_p(amax_bits.data_ptr()),
            ctypes.c_longlong(xd.numel())), "quant_fp8")
        return q, scale
    amax = x.detach().abs().amax().float().clamp_(min=2.0 ** -20)
    scale = amax / 448.0
    q = (x.float() / scale).to(torch.float8_e4m3fn)
    return q, scale


def _quant_fp8_rowwise(x: torch.Tensor):
    """Per-ROW pow2 quantization for the hardware-scaled MX MFMA path.

    Returns (q, e8, ratio, s_ref): x_row ≈ q_row · 2^(e8_row−127);
    ratio_row = 2^(e8_row − e_max) ≤ 1 (the backward slab fold), and
    s_ref = 2^(e_max−127) (the per-tensor factor the GEMM scale carries).
    """
    lib = _require_lib()
    xd = x.detach().contiguous()
    b, d = xd.shape
    q = torch.empty((b, d), device=xd.device, dtype=torch.float8_e4m3fn)
    e8 = torch.empty(b, device=xd.device, dtype=torch.uint8)
    emax = torch.zeros(1, device=xd.device, dtype=torch.int32)
    stream = torch.cuda.current_stream(xd.device).cuda_stream
    _check(lib.quant_fp8_rowwise_bf16(
        ctypes.c_void_p(stream), ctypes.c_void_p(xd.data_ptr()),
        ctypes.c_void_p(q.data_ptr()), ctypes.c_void_p(e8.data_ptr()),
        ctypes.c_void_p(emax.data_ptr()), b, d), "quant_fp8_rowwise")
    emax_f = emax[0].float()
    ratio = torch.exp2(e8.float() - emax_f)
    s_ref = torch.exp2(emax_f - 127.0)
    return q, e8, ratio, s_ref


def rowwise_ok(b: int, n: int, d: int) -> bool:
    """Row-wise fp8 policy gate: needs the saved-g mm8-aligned shapes (the
    slab folds assume the fp8 GEMM path) and SIGLIP_FP8_ROWWISE=1.

    OFF by default: it costs ~14%% of fp8 step throughput (4.84 vs
    4.24 ms at B=32k — epilogue ratio folds + the row-wise quant pass)
    and the loss contract assumes unit-norm rows, where per-tensor scales
    lose nothing.  Turn it on for embeddings with per-row dynamic range
    beyond e4m3's ~2^18 (see test_rowwise_quant_keeps_per_row_precision)."""
    return (os.environ.get("SIGLIP_FP8_ROWWISE", "0") == "1"
            and b % 16 == 0 and n % 16 == 0 and d % 16 == 0
            and hasattr(torch, "_scaled_mm"))


def quantize_fp8_rowwise_pair(zimg: torch.Tensor, ztxt: torch.Tensor):
    """Row-wise qcache: an 8-tuple (zi_q, zi_e8, zi_ratio, si_ref,
    zt_q, zt_e8, zt_ratio, st_ref) — distinguished from the per-tensor
    4-tuple by length."""
    return _quant_fp8_rowwise(zimg) + _quant_fp8_rowwise(ztxt)


def quantize_fp8_pair(zimg: torch.Tensor, ztxt: torch.Tensor):
    """Quantize both embedding tensors once; pass the result as ``qcache`` to
    both :func:`siglip_fwd` and :func:`siglip_bwd` so an fwd+bwd step pays a
    single quantization pass."""
    zi_q, si = _quant_fp8(zimg)
    zt_q, st = _quant_fp8(ztxt)
    return zi_q, si, zt_q, st


def siglip_fwd(zimg: torch.Tensor, ztxt: torch.Tensor, t_prime: torch.Tensor,
               bias: torch.Tensor, diag_offset: Optional[int],
               quant: str = "bf16", qcache=None) -> torch.Tensor:
    lib = _require_lib()
    _validate(zimg, ztxt, quant)
    if quant == "mixed":
        quant = "bf16"   # mixed = bf16 logits; fp8 applies to backward only
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    bp = _prep_scalar(bias, dev)
    extra = ()
    if quant == "fp8":
        if qcache is not None and len(qcache) == 8:
            # row-wise policy: hardware dequant, tp stays unmodified
            zi_q, zi_e8, _, _, zt_q, zt_e8, _, _ = qcache
            extra = (ctypes.c_void_p(zi_e8.data_ptr()),
                     ctypes.c_void_p(zt_e8.data_ptr()))
        else:
            zi_q, si, zt_q, st = (qcache if qcache is not None
                                  else quantize_fp8_pair(zimg, ztxt))
            tp = tp + si.log() + st.log()
            extra = (None, None)
        zi_ptr, zt_ptr = zi_q.data_ptr(), zt_q.data_ptr()
        fn = lib.siglip_fwd_fp8
    else:
        zi_ptr, zt_ptr = zimg.data_ptr(), ztxt.data_ptr()
        fn = lib.siglip_fwd_bf16
    buf = _out_buf(dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    diag = _DIAG_NONE if diag_offset is None else int(diag_offset)
    _check(fn(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(zi_ptr), ctypes.c_void_p(zt_ptr),
        ctypes.c_void_p(tp.data_ptr()), ctypes.c_void_p(bp.data_ptr()),
        ctypes.c_void_p(buf.data_ptr()), b, n, d, diag,
        _kernel_flags(default_gm="8"), *extra),
        "siglip_fwd")
    return buf[:, 0].sum()


def siglip_bwd(zimg: torch.Tensor, ztxt: torch.Tensor, t_prime: torch.Tensor,
               bias: torch.Tensor, diag_offset: Optional[int],
               grad_output: torch.Tensor, col_chunk: Optional[int],
               quant: str = "bf16", on_dztxt=None, qcache=None):
    """Returns (dzimg, dztxt, dt_prime, dbias).

    Per column slab: the fused kernel recomputes logit tiles (MFMA) and
    writes g = dL/d(logit pre-scale); then dzimg += g @ ztxt_slab and
    dztxt_slab = gᵀ @ zimg on the GEMM libraries (bf16 rocBLAS, or fp8
    hipBLASLt when the policy provides e4m3 slabs).  ``on_dztxt(dztxt)`` is
    invoked as soon as the text gradient exists so distributed callers can
    overlap their reduce-scatter with the remaining image-gradient GEMM.
    """
    lib = _require_lib()
    _validate(zimg, ztxt, quant)
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    bp = _prep_scalar(bias, dev)
    t_true = tp.exp()
    if quant in ("fp8", "mixed"):
        # Reuse the forward's quantization when provided (identical inputs
        # give identical amax, so recomputing is equivalent but wasteful);
        # the g kernel sees t_eff so its logits match the forward's.
        # (The recompute path is per-tensor only — row-wise qcaches ride
        # the saved-g path, ops.siglip_bwd_from_g.)
        if qcache is not None and len(qcache) == 8:
            raise RuntimeError(
                "row-wise fp8 qcache requires the saved-g backward")
        zi_q, si, zt_q, st = (qcache if qcache is not None
                              else quantize_fp8_pair(zimg, ztxt))
        zi_g, zt_g = zi_q, zt_q           # fp8 GEMM operands
        if quant == "fp8":
            # fp8 logits: kernel reads quantized inputs with t_eff.
            tp_k = tp + si.log() + st.log()
            g_fn = lib.siglip_bwd_g_fp8
            zi_k, zt_k = zi_q, zt_q
        else:
            # mixed: bf16 logits recompute; quantized inputs feed only the
            # fp8 gradient GEMMs below.
            tp_k = tp
            g_fn = lib.siglip_bwd_g_mixed
            zi_k, zt_k = zimg, ztxt
    else:
        tp_k = tp
        g_fn = lib.siglip_bwd_g_bf16
        zi_k, zt_k = zimg, ztxt
    scal = _out_buf(dev)
    stream = torch.cuda.current_stream(dev).cuda_stream

    # Column slab sizing: the kernel's g-store addressing is 32-bit, so
    # b * slab * esz bytes must stay below 2^32; beyond that (and to bound
    # workspace at huge n) we chunk.
    fp8g = quant in ("fp8", "mixed")   # g slabs are e4m3 (×448)
    g_esz = 1 if fp8g else 2
    g_dtype = torch.float8_e4m3fn if fp8g else torch.bfloat16
    step = col_chunk if col_chunk and col_chunk > 0 else n
    while step > 256 and (b * step * g_esz) >= 2 ** 32:
        step //= 2
    # Floor of 256 columns per slab: below that the kernel grid degenerates.
    # The 32-bit invariant must survive the floor — at bf16 that means
    # b < 2^32/(256·2) = 8.4M rows, far beyond any real shard.
    step = max(step, 256)
    if (b * step * g_esz) >= 2 ** 32:
        raise RuntimeError(
            f"batch {b} too large for the 32-bit g-slab addressing even at "
            f"the 256-column slab floor; shard the batch")

    go = grad_output.detach().reshape(()).to(device=dev, dtype=torch.float32)
    scale = go * t_true   # gradient GEMMs run against the original bf16

    # fp8 gradient GEMMs (hipBLASLt via torch._scaled_mm, measured 2.6× the
    # bf16 matmul): the kernel's g slab is e4m3 at a fixed ×448 scale, the
    # quantized embeddings carry per-tensor scales — all folded into
    # _scaled_mm's scale args so no extra elementwise passes run.  Shapes
    # must be 16-aligned; otherwise dequantize and use rocBLAS.
    use_mm8 = (fp8g and b % 16 == 0 and n % 16 == 0
               and d % 16 == 0 and hasattr(torch, "_scaled_mm"))
    if fp8g and b % 4 != 0:   # packed gt stores need b%4
        raise RuntimeError(f"{quant} backward requires batch % 4 == 0")

    _one = torch.ones((), device=dev)

    def mm8(a8, b8_rowmajor, s_ab):
        """a8 (m, k) @ b8 (k, d) → bf16; mat2 re-laid column-major as
        _scaled_mm requires (k ≤ a few thousand rows — cheap copy)."""
        b_cm = b8_rowmajor.t().contiguous().t()   # column-major (k, d)
        return torch._scaled_mm(a8, b_cm, scale_a=s_ab, scale_b=_one,
                                out_dtype=torch.bfloat16)

    def run_g(j0, j1, g_slab, diag, gt_slab=None):
        zt_slab = zt_k[j0:j1]
        if fp8g:
            extra = ((None, None, None, None) if quant == "fp8" else ())
            _check(g_fn(
                ctypes.c_void_p(stream),
                ctypes.c_void_p(zi_k.data_ptr()),
                ctypes.c_void_p(zt_slab.data_ptr()),
                ctypes.c_void_p(tp_k.data_ptr()),
                ctypes.c_void_p(bp.data_ptr()),
                ctypes.c_void_p(g_slab.data_ptr()),
                ctypes.c_void_p(gt_slab.data_ptr()),
                ctypes.c_void_p(scal.data_ptr()),
                b, j1 - j0, d, diag, _kernel_flags(), *extra),
                "siglip_bwd_g_fp8/mixed")
        else:
            _check(g_fn(
                ctypes.c_void_p(stream),
                ctypes.c_void_p(zi_k.data_ptr()),
                ctypes.c_void_p(zt_slab.data_ptr()),
                ctypes.c_void_p(tp_k.data_ptr()),
                ctypes.c_void_p(bp.data_ptr()),
                ctypes.c_void_p(g_slab.data_ptr()),
                ctypes.c_void_p(scal.data_ptr()),
                b, j1 - j0, d, diag, _kernel_flags()),
                "siglip_bwd_g_bf16")

    if step >= n:
        # Single-slab fast path: no fp32 accumulation round trips.  dztxt is
        # produced FIRST so the caller's on_dztxt hook (e.g. an async RCCL
        # reduce-scatter) overlaps with the dzimg GEMM below.
        g = torch.empty((b, n), device=dev, dtype=g_dtype)
        gt = (torch.empty((n, b), device=dev, dtype=g_dtype)
              if fp8g else None)
        run_g(0, n, g, _DIAG_NONE if diag_offset is None else int(diag_offset),
              gt)
        if use_mm8:
            s_t = ((scale / 448.0) * st).reshape(())
            s_i = ((scale / 448.0) * si).reshape(())
            dztxt = mm8(gt, zi_g, s_i)
            if on_dztxt is not None:
                on_dztxt(dztxt)
            dzimg = mm8(g, zt_g, s_t)
        else:
            if fp8g:
                g = g.to(torch.bfloat16) * (1.0 / 448.0)
            dztxt = ((g.T @ zimg) * scale).to(ztxt.dtype)
            if on_dztxt is not None:
                on_dztxt(dztxt)
            dzimg = (g @ ztxt) * scale
    else:
        dzimg_acc = torch.zeros((b, d), device=dev, dtype=torch.float32)
        dztxt = torch.empty((n, d), device=dev, dtype=torch.bfloat16)
        g_buf = torch.empty((b, step), device=dev, dtype=g_dtype)
        gt_buf = (torch.empty((step, b), device=dev, dtype=g_dtype)
                  if fp8g else None)
        for j0 in range(0, n, step):
            j1 = min(j0 + step, n)
            c = j1 - j0
            g = g_buf if c == step else torch.empty(
                (b, c), device=dev, dtype=g_dtype)
            gt = None
            if fp8g:
                gt = gt_buf if c == step else torch.empty(
                    (c, b), device=dev, dtype=g_dtype)
            diag = _DIAG_NONE if diag_offset is None else int(diag_offset) - j0
            run_g(j0, j1, g, diag, gt)
            if use_mm8 and c % 16 == 0:
                s_t = ((scale / 448.0) * st).reshape(())
                s_i = ((scale / 448.0) * si).reshape(())
                dzimg_acc += mm8(g, zt_g[j0:j1], s_t).float()
                dztxt[j0:j1] = mm8(gt, zi_g, s_i)
            else:
                g16 = (g.to(torch.bfloat16) * (1.0 / 448.0)
                       ) if fp8g else g
                # scale applied per chunk so mixed mm8/fallback chunks agree
                dzimg_acc += ((g16 @ ztxt[j0:j1]) * scale).float()
                dztxt[j0:j1] = (g16.T @ zimg) * scale
        if on_dztxt is not None:
            on_dztxt(dztxt)
        dzimg = dzimg_acc

    dzimg = dzimg.to(zimg.dtype)
    dztxt = dztxt.to(ztxt.dtype)
    # dt' = go · Σ g·(z−bias) = go · t_eff · Σ g·dot_q  (t_eff ≡ t for bf16).
    t_eff = tp_k.exp()
    sv = reduce_out3(scal)
    dt_prime = (sv[1] * go * t_eff).to(t_prime.dtype).reshape(t_prime.shape)
    dbias = (sv[2] * go).to(bias.dtype).reshape(bias.shape)
    return dzimg, dztxt, dt_prime, dbias


# ---------------------------------------------------------------------------
# fwd+g ("saved-g") path: one kernel emits loss + g slab + scalar partials,
# so backward is pure GEMMs — the training step never computes the logits
# GEMM twice.  Used whenever the slab fits comfortably in HBM (it is 2 GiB
# at the B=32k bf16 headline config, against 288 GB); the recompute kernels
# above remain for the chunked huge-batch path (BASELINE config 3).
# ---------------------------------------------------------------------------


def save_g_enabled(b: int, n: int, quant: str) -> bool:
    """Policy: is the fwd+g saved-slab path usable/worth it for this block?

    SIGLIP_SAVE_G = auto (default) | 0 (always recompute) | 1 (force when
    addressable); SIGLIP_SAVE_G_MAX_BYTES bounds the slab workspace in auto
    mode (default 6 GiB — g plus, for fp8/mixed, its transpose).
    """
    env = os.environ.get("SIGLIP_SAVE_G", "auto")
    if env == "0":
        return False
    esz = 1 if quant in ("fp8", "mixed") else 2
    if b * n * esz >= 2 ** 32:        # kernel's 32-bit slab addressing
        return False
    if env == "1":
        return True
    need = b * n * esz * (2 if esz == 1 else 1)
    cap = int(os.environ.get("SIGLIP_SAVE_G_MAX_BYTES", str(6 * 2 ** 30)))
    return need <= cap


def save_g_banded_enabled(b: int, n: int, quant: str) -> bool:
    """Column-banded saved-g policy: when ONE slab exceeds the kernel's
    32-bit addressing but the full (b, n) g fits HBM comfortably (it is
    32 GB at 131k², against 288 GB), forward emits g in column bands and
    backward is still pure GEMMs — no logits recompute.  bf16 only (the
    fp8 slabs come in pairs and ride the single-slab path);
    SIGLIP_SAVE_G_BANDED=0 disables, SIGLIP_SAVE_G=0 disables all saving."""
    if quant != "bf16":
        return False
    if os.environ.get("SIGLIP_SAVE_G", "auto") == "0":
        return False
    if os.environ.get("SIGLIP_SAVE_G_BANDED", "auto") == "0":
        return False
    if not torch.cuda.is_available():
        return False
    total = b * n * 2
    try:
        free, _ = torch.cuda.mem_get_info()
    except Exception:  # pragma: no cover
        return False
    return total <= free * 0.5


def banded_col_step(b: int, esz: int = 2) -> int:
    """Widest 256-aligned column band whose slab stays under the kernel's
    32-bit store addressing (SIGLIP_BANDED_STEP overrides, for tests)."""
    env = os.environ.get("SIGLIP_BANDED_STEP")
    if env:
        return max(256, int(env))
    step = (2 ** 32 - 1) // (b * esz)
    return max(256, (step // 256) * 256)


def scaled_mm8(a8: torch.Tensor, b8_rowmajor: torch.Tensor,
               scale: torch.Tensor) -> torch.Tensor:
    """fp8 GEMM a8 (m, k) @ b8 (k, d) → bf16 via hipBLASLt
    (``torch._scaled_mm``; mat2 re-laid column-major as it requires — k is at
    most a few thousand rows, a cheap copy)."""
    one = torch.ones((), device=a8.device)
    b_cm = b8_rowmajor.t().contiguous().t()
    return torch._scaled_mm(a8, b_cm, scale_a=scale.reshape(()), scale_b=one,
                            out_dtype=torch.bfloat16)


def siglip_fwd_g(zimg: torch.Tensor, ztxt: torch.Tensor,
                 t_prime: torch.Tensor, bias: torch.Tensor,
                 diag_offset: Optional[int], quant: str = "bf16",
                 qcache=None, g_slab: Optional[torch.Tensor] = None,
                 gt_slab: Optional[torch.Tensor] = None, col0: int = 0,
                 out3: Optional[torch.Tensor] = None):
    """Fused forward that also emits the g slab and both scalar partials.

    Returns ``(out3, g_slab, gt_slab)`` where ``out3`` is the RAW (8, 32)
    per-XCD scalar buffer (atomically accumulated — pass the same buffer
    across chunk calls to sum them; reduce with :func:`reduce_out3` →
    ``[loss, Σ g·dot, Σ g]``), ``g_slab`` is the ``(b, n)`` dL/dlogit
    slab (bf16, or e4m3 ×448 for fp8/mixed along with its ``(n, b)``
    transpose ``gt_slab``).  When ``g_slab`` is supplied the chunk is written
    at column offset ``col0`` with the slab's width as row stride — the ring
    strategy assembles one ``(b, W·b)`` slab chunk by chunk.
    """
    lib = _require_lib()
    _validate(zimg, ztxt, quant)
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    bp = _prep_scalar(bias, dev)
    fp8g = quant in ("fp8", "mixed")
    if fp8g and b % 4 != 0:
        raise RuntimeError(f"{quant} fwd+g requires batch % 4 == 0")
    esz = 1 if fp8g else 2
    g_dtype = torch.float8_e4m3fn if fp8g else torch.bfloat16
    if g_slab is None:
        g_slab = torch.empty((b, n), device=dev, dtype=g_dtype)
    ldg = g_slab.shape[1]
    g_ptr = g_slab.data_ptr() + col0 * esz
    gt_ptr = 0
    if fp8g:
        if gt_slab is None:
            gt_slab = torch.empty((n, b), device=dev, dtype=g_dtype)
        gt_ptr = gt_slab.data_ptr() + col0 * b * esz
    if b * ldg * esz >= 2 ** 32:
        raise RuntimeError(
            "g slab exceeds the kernel's 32-bit addressing; use the "
            "recompute path (save_g_enabled would have said no)")
    if out3 is None:
        out3 = _out_buf(dev)
    rs_extra = ()
    if quant == "fp8":
        if qcache is not None and len(qcache) == 8:
            # row-wise: e8m0 scales feed the MFMA, ratios feed the slab
            # folds; tp stays unmodified (acc is the true dot).
            zi_q, zi_e8, zi_rat, _, zt_q, zt_e8, zt_rat, _ = qcache
            tp_k = tp
            rs_extra = (ctypes.c_void_p(zi_e8.data_ptr()),
                        ctypes.c_void_p(zt_e8.data_ptr()),
                        ctypes.c_void_p(zi_rat.data_ptr()),
                        ctypes.c_void_p(zt_rat.data_ptr()))
        else:
            zi_q, si, zt_q, st = (qcache if qcache is not None
                                  else quantize_fp8_pair(zimg, ztxt))
            tp_k = tp + si.log() + st.log()
            rs_extra = (None, None, None, None)
        fn = lib.siglip_fwdg_fp8
        zi_ptr, zt_ptr = zi_q.data_ptr(), zt_q.data_ptr()
    else:
        tp_k = tp
        fn = lib.siglip_fwdg_mixed if quant == "mixed" else lib.siglip_fwdg_bf16
        zi_ptr, zt_ptr = zimg.data_ptr(), ztxt.data_ptr()
    stream = torch.cuda.current_stream(dev).cuda_stream
    diag = _DIAG_NONE if diag_offset is None else int(diag_offset)
    if fp8g:
        rc = fn(ctypes.c_void_p(stream), ctypes.c_void_p(zi_ptr),
                ctypes.c_void_p(zt_ptr), ctypes.c_void_p(tp_k.data_ptr()),
                ctypes.c_void_p(bp.data_ptr()),
                ctypes.c_void_p(out3.data_ptr()), ctypes.c_void_p(g_ptr),
                ctypes.c_void_p(gt_ptr), b, n, d, ldg, diag, _kernel_flags(),
                *rs_extra)
    else:
        rc = fn(ctypes.c_void_p(stream), ctypes.c_void_p(zi_ptr),
                ctypes.c_void_p(zt_ptr), ctypes.c_void_p(tp_k.data_ptr()),
                ctypes.c_void_p(bp.data_ptr()),
                ctypes.c_void_p(out3.data_ptr()), ctypes.c_void_p(g_ptr),
                b, n, d, ldg, diag, _kernel_flags())
    _check(rc, "siglip_fwdg")
    return out3, g_slab, gt_slab if fp8g else None


class _FusedL2Normalize(torch.autograd.Function):
    """Row-wise L2 normalize via the single-pass HIP kernels (bf16, fp32
    math; torch ``F.normalize(dim=-1)`` semantics) — replaces the ~10-kernel
    autograd chain on the towers' hot path."""

    @staticmethod
    def forward(ctx, x, eps):
        lib = _require_lib()
        x = x.contiguous()
        b, d = x.shape
        y = torch.empty_like(x)
        rn = torch.empty(b, device=x.device, dtype=torch.float32)
        stream = torch.cuda.current_stream(x.device).cuda_stream
        _check(lib.l2norm_fwd_bf16(
            ctypes.c_void_p(stream), ctypes.c_void_p(x.data_ptr()),
            ctypes.c_void_p(y.data_ptr()), ctypes.c_void_p(rn.data_ptr()),
            b, d, ctypes.c_float(eps)), "l2norm_fwd")
        ctx.save_for_backward(y, rn)
        return y

    @staticmethod
    def backward(ctx, dy):
        lib = _require_lib()
        y, rn = ctx.saved_tensors
        dy = dy.contiguous()
        b, d = y.shape
        dx = torch.empty_like(y)
        stream = torch.cuda.current_stream(y.device).cuda_stream
        _check(lib.l2norm_bwd_bf16(
            ctypes.c_void_p(stream), ctypes.c_void_p(dy.data_ptr()),
            ctypes.c_void_p(y.data_ptr()), ctypes.c_void_p(rn.data_ptr()),
            ctypes.c_void_p(dx.data_ptr()), b, d), "l2norm_bwd")
        return dx, None


def l2_normalize(x: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    """L2-normalize rows.  GPU bf16 2-D inputs take the fused kernel pair;
    everything else falls back to ``F.normalize`` (identical semantics)."""
    if (x.is_cuda and x.dtype == torch.bfloat16 and x.dim() == 2
            and x.shape[1] % 2 == 0 and extension_available()):
        return _FusedL2Normalize.apply(x, eps)
    import torch.nn.functional as F
    return F.normalize(x, dim=-1, eps=eps)


def siglip_bwd_from_g(zimg: torch.Tensor, ztxt: torch.Tensor,
                      t_prime: torch.Tensor, bias: torch.Tensor,
                      grad_output: torch.Tensor, out3: torch.Tensor,
                      g: torch.Tensor, gt: Optional[torch.Tensor],
                      quant: str = "bf16", qcache=None, on_dztxt=None):
    """Backward from the fwd+g saved slab: two gradient GEMMs plus the scalar
    grads — no logits recompute.  ``out3`` is the REDUCED 3-vector
    (:func:`reduce_out3`).  Same return contract as :func:`siglip_bwd`.

    ``on_dztxt(dztxt)`` fires as soon as the text gradient exists so a
    distributed caller overlaps its reduce-scatter with the dzimg GEMM.
    """
    b, d = zimg.shape
    n = ztxt.shape[0]
    dev = zimg.device
    tp = _prep_scalar(t_prime, dev)
    t_true = tp.exp()
    go = grad_output.detach().reshape(()).to(device=dev, dtype=torch.float32)
    scale = go * t_true
    fp8g = quant in ("fp8", "mixed")
    if fp8g and qcache is not None and len(qcache) == 8:
        # row-wise policy: the slabs already fold the per-row ratios; the
        # GEMM scale carries only the tensor-max factor, and dt' needs no
        # scale correction (the MFMA accumulated the true dot).
        zi_q, _, _, si_ref, zt_q, _, _, st_ref = qcache
        dztxt = scaled_mm8(gt, zi_q, (scale / 448.0) * si_ref)
        if on_dztxt is not None:
            on_dztxt(dztxt)
        dzimg = scaled_mm8(g, zt_q, (scale / 448.0) * st_ref)
        t_eff = t_true
    elif fp8g:
        zi_q, si, zt_q, st = qcache
        use_mm8 = (b % 16 == 0 and n % 16 == 0 and d % 16 == 0
                   and hasattr(torch, "_scaled_mm"))
        if use_mm8:
            dztxt = scaled_mm8(gt, zi_q, (scale / 448.0) * si)
            if on_dztxt is not None:
                on_dztxt(dztxt)
            dzimg = scaled_mm8(g, zt_q, (scale / 448.0) * st)
        else:
            g16 = g.to(torch.bfloat16) * (1.0 / 448.0)
            dztxt = (g16.T @ zimg) * scale
            if on_dztxt is not None:
                on_dztxt(dztxt)
            dzimg = (g16 @ ztxt) * scale
        t_eff = t_true * si * st if quant == "fp8" else t_true
    else:
        # Scale in the GEMM output's own dtype: a fp32 0-d multiplier
        # promotes the whole (n, d) product to fp32 (two extra full-tensor
        # passes for the round trip); the bf16 scalar rounding (~0.4%) is
        # below the bf16 grad noise floor.
        scale_c = scale.to(g.dtype)
        dztxt = (g.T @ zimg) * scale_c
        if on_dztxt is not None:
            on_dztxt(dztxt)
        dzimg = (g @ ztxt) * scale_c
        t_eff = t_true
    dt_prime = (out3[1] * go * t_eff).to(t_prime.dtype).reshape(t_prime.shape)
    dbias = (out3[2] * go).to(bias.dtype).reshape(bias.shape)
    return (dzimg.to(zimg.dtype), dztxt.to(ztxt.dtype), dt_prime, dbias)
