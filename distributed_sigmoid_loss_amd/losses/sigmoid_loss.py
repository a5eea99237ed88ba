"""Distributed SigLIP loss modules.

Two public modules, mirroring the reference repo's API surface:

- :class:`DistributedSigmoidLoss` — module-owned learnable ``t_prime`` (init
  ``log 10``) and ``bias`` (init ``−10``); forward takes the *local* image and
  text embedding shards and returns ``Σ loss / gpu_batch_size``.  Behavioral
  parity with reference ``distributed_sigmoid_loss.py:8-48``.  Strategy is
  selectable: ``"all_gather"`` (differentiable all-gather, the reference's
  scheme) or ``"ring"`` (pipelined P2P ring with comm/compute overlap — the
  MI355X performance path).
- :class:`SigLipLoss` — caller-owned ``logit_scale``/``logit_bias`` passed to
  ``forward`` as arguments; ring strategy via the autograd neighbour-exchange
  primitives.  Behavioral parity with reference
  ``rwightman_sigmoid_loss.py:12-124`` (including ``bidir`` and the
  ``output_dict`` return form).

Unlike the reference (which builds labels on the default CPU device,
``distributed_sigmoid_loss.py:28-30``), everything here is device- and
dtype-correct: labels are an ``i == j + offset`` index predicate, never a
materialized tensor on the hot path.
"""

from __future__ import annotations

import math
import os
from typing import Optional

import torch
import torch.nn as nn
import torch.distributed as dist

from .functional import (
    sigmoid_contrastive_loss,
    chunk_loss_fwd,
    chunk_loss_bwd,
)
from ..parallel.collectives import all_gather_with_grad, _backend_is_gloo
from ..parallel.ring import (
    neighbour_exchange_with_grad,
    neighbour_exchange_bidir_with_grad,
    neighbour_exchange_start,
    neighbour_exchange_bidir_start,
    quantized_exchange_start,
    quantized_exchange_bidir_start,
)
from ..utils.profiling import roctx_range, HopStats


import contextlib


@contextlib.contextmanager
def _hop_span(label):
    # roctx marker for external profilers + CUDA-event pair for the
    # in-process per-hop stats (SIGLIP_HOP_STATS=1 / bench --csv).
    with roctx_range(label), HopStats.record(label):
        yield


def _world_and_rank(group=None):
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size(group), dist.get_rank(group)
    return 1, 0


class _RingAllGatherLoss(torch.autograd.Function):
    """Fused distributed loss: ring-pipelined forward, reduce-scatter backward.

    Forward walks the text shards around a P2P ring — unidirectional, or
    bidirectional with two xGMI links per hop and ⌈(W−1)/2⌉ hops; the
    exchange of the next chunk is posted *before* the loss kernel for the
    current one runs, so on RCCL the wire time hides under the MFMA compute
    (the reference serializes these — ``distributed_utils.py:25-26``).  All
    received shards are kept (O(W·b·d), trivial against 288 GB HBM3E) so
    backward never re-communicates embeddings.  Each chunk's fwd+g kernel
    writes its dL/dlogit block into one (b, W·b) slab (saved-g, the default)
    so backward is pure GEMMs; the recompute path remains for huge batches.

    Backward produces the full text-gradient block and issues ONE
    ``reduce_scatter_tensor(SUM)`` — the algebraic collapse of the
    reference's W−1 reversed autograd ring hops
    (``distributed_utils.py:74-77``) into a single RCCL collective —
    launched async from the ``on_dztxt`` hook so its wire time overlaps the
    remaining image-gradient GEMM.

    Gradient semantics match the differentiable all-gather strategy exactly:
    raw per-rank grads already carry full cross-rank contributions (what the
    reference's strategy-equivalence oracle asserts,
    ``test_sigmoid_loss_variants.py:112-113``).
    """

    @staticmethod
    def forward(ctx, zimg, ztxt, t_prime, bias, group, col_chunk, impl,
                quant, want_grad, bidir=False):
        from .. import ops as _ops
        world, rank = _world_and_rank(group)
        zimg = zimg.contiguous()
        ztxt = ztxt.contiguous()
        b_img = zimg.shape[0]
        b_txt = ztxt.shape[0]

        chunks = [None] * world          # indexed by source rank
        chunks[rank] = ztxt

        on_gpu = zimg.is_cuda and impl != "torch"
        # saved-g: each chunk's fwd+g kernel writes its dL/dlogit block
        # straight into one (b, W·b) slab (column offset = source rank), so
        # backward is pure GEMMs — no logits recompute anywhere in the step.
        # want_grad comes from the caller (grad mode is off inside
        # Function.forward).
        save_g = (on_gpu and col_chunk is None and want_grad
                  and _ops.extension_available()
                  and _ops.save_g_enabled(b_img, world * b_txt, quant))

        # fp8: quantize BOTH own shards ONCE and ship e4m3 + scale over the
        # wire — half the xGMI bytes per hop versus bf16, no per-chunk
        # quantize kernels, and every rank computes with the exact
        # quantized values the owner produced.  Per-chunk scales are SAVED
        # for backward so fwd and bwd see identical quantized values (the
        # round-1 version requantized the concatenated block with one
        # scale — fwd/bwd loss surfaces differed at W>1).
        fp8_wire = quant == "fp8" and world > 1
        use_q = quant == "fp8" and on_gpu
        qcaches = [None] * world        # per-source (zi_q, si, zt_q, st)
        if quant == "fp8":
            zi_q, si = _ops._quant_fp8(zimg)
            zt_q0, st0 = _ops._quant_fp8(ztxt)

            def dequant(q8, sc):
                return (q8.to(torch.float32) * sc).to(zimg.dtype)

            def qc_of(q8, sc):
                return (zi_q, si, q8, sc.reshape(()))
        else:
            zt_q0 = st0 = None

        def qc_for(zt_chunk):
            # own-rank / non-wire chunks only (wire chunks carry their
            # quantization with them)
            if quant != "fp8":
                return None
            if zt_chunk is ztxt:
                return qc_of(zt_q0, st0)
            zt_q, st = _ops._quant_fp8(zt_chunk)
            return (zi_q, si, zt_q, st)

        g_slab = gt_slab = out3 = None
        g_chunks = [None] * world       # fp8: per-chunk g (scales differ)
        out3s = [None] * world
        if save_g:
            esz = 1 if quant in ("fp8", "mixed") else 2
            g_dtype = (torch.float8_e4m3fn if esz == 1 else torch.bfloat16)
            if quant != "fp8":
                g_slab = torch.empty((b_img, world * b_txt),
                                     device=zimg.device, dtype=g_dtype)
                out3 = _ops._out_buf(zimg.device)
            if quant in ("fp8", "mixed"):
                gt_slab = torch.empty((world * b_txt, b_img),
                                      device=zimg.device, dtype=g_dtype)

        def chunk_fwd(zt_chunk, src, diag, qc=None):
            if qc is None:
                qc = qc_for(zt_chunk)
            qcaches[src] = qc
            if not save_g:
                return chunk_loss_fwd(zimg, zt_chunk, t_prime, bias,
                                      diag_offset=diag, col_chunk=col_chunk,
                                      impl=impl, quant=quant, qcache=qc)
            if quant == "fp8":
                o3, g_c, _ = _ops.siglip_fwd_g(
                    zimg, zt_chunk, t_prime, bias, diag, quant=quant,
                    qcache=qc, gt_slab=gt_slab, col0=src * b_txt)
                g_chunks[src] = g_c
                out3s[src] = o3
                return None
            _ops.siglip_fwd_g(
                zimg, zt_chunk, t_prime, bias, diag, quant=quant,
                g_slab=g_slab, gt_slab=gt_slab, col0=src * b_txt, out3=out3)
            return None   # shared out3 accumulates across chunks

        # Wire payload abstraction: bf16/mixed ship the chunk tensor; fp8
        # ships its (e4m3-as-uint8, fp32 scale) pair.  unpack() returns the
        # compute-dtype chunk plus the per-chunk qcache.
        if fp8_wire:
            def wire_uni(payload):
                return quantized_exchange_start(left, right, payload[0],
                                                payload[1], group=group)

            def wire_bidir(pl, pr):
                return quantized_exchange_bidir_start(
                    left, right, pl[0], pl[1], pr[0], pr[1], group=group)

            def unpack(recvs):
                q8 = recvs[0].view(torch.float8_e4m3fn)
                sc = recvs[1]
                return dequant(q8, sc), qc_of(q8, sc), (q8, sc)

            own_payload = (zt_q0, st0.reshape(1).float())
        else:
            def wire_uni(payload):
                return neighbour_exchange_start(left, right, payload,
                                                group=group)

            def wire_bidir(pl, pr):
                return neighbour_exchange_bidir_start(left, right, pl, pr,
                                                      group=group)

            def unpack(recvs):
                return recvs[0], None, recvs[0]

            own_payload = ztxt

        use_bidir = bidir and world > 2
        if world > 1:
            left = (rank - 1 + world) % world
            right = (rank + 1) % world
            if group is not None:
                # P2POp peers are GLOBAL ranks even when a group is given;
                # rank/left/right above are group-local.  Identity for the
                # default group, required for subgroups like {1, 2}.
                left = dist.get_global_rank(group, left)
                right = dist.get_global_rank(group, right)
            # Post hop 1 before any compute: its wire time hides under the
            # local-block kernel below.  The bidirectional variant drives
            # TWO xGMI point-to-point links per hop (reference
            # distributed_utils.py:30-62 at the perf layer), halving the hop
            # count to ⌈(W−1)/2⌉ — at N=8 per-rank b=4096 the ring is
            # wire-bound, so hop count and bytes per hop are the levers.
            if use_bidir:
                handle = wire_bidir(own_payload, own_payload)
            else:
                handle = wire_uni(own_payload)

        loss = chunk_fwd(ztxt, rank, 0)

        def add_part(part):
            nonlocal loss
            if part is not None:
                loss = loss + part

        def take(recvs, src):
            chunk, qc, payload = unpack(recvs)
            chunks[src] = chunk
            return chunk, qc, payload

        if world > 1 and use_bidir:
            nb, rem = divmod(world - 1, 2)
            npl = 2 if fp8_wire else 1    # tensors per direction on the wire
            for r in range(1, nb + 1):
                with _hop_span(f"ring_bhop{r}_wait"):
                    recvs = handle.wait()
                src_r = (rank + r) % world
                src_l = (rank - r + world) % world
                from_right, qc_r, pl_r = take(recvs[:npl], src_r)
                from_left, qc_l, pl_l = take(recvs[npl:], src_l)
                if r < nb:
                    handle = wire_bidir(pl_r, pl_l)
                elif rem:
                    # one unidirectional remainder hop: forward the
                    # rightward-traveling stream (last received from left)
                    handle = wire_uni(pl_l)
                with _hop_span(f"ring_bchunk{r}_loss"):
                    add_part(chunk_fwd(from_right, src_r, None, qc_r))
                    add_part(chunk_fwd(from_left, src_l, None, qc_l))
            if rem:
                with _hop_span("ring_rem_wait"):
                    recvs = handle.wait()
                src = (rank - nb - 1 + world) % world
                recv, qc, _ = take(recvs, src)
                add_part(chunk_fwd(recv, src, None, qc))
        elif world > 1:
            for hop in range(1, world):
                # Post hop k+1 before computing on hop k's data.  roctx
                # ranges label the hops for rocprofv3/torch.profiler traces
                # (per-hop comm visibility).
                with _hop_span(f"ring_hop{hop}_wait"):
                    recvs = handle.wait()
                src = (rank - hop + world) % world
                recv, qc, payload = take(recvs, src)
                if hop < world - 1:
                    handle = wire_uni(payload)
                with _hop_span(f"ring_chunk{hop}_loss"):
                    add_part(chunk_fwd(recv, src, None, qc))

        if save_g:
            # Reduce the per-XCD scalar buffers once, after every chunk's
            # kernel has accumulated into them.
            if quant == "fp8":
                out3s = [_ops.reduce_out3(o) for o in out3s]
                loss = sum(o[0] for o in out3s).clone()
            else:
                out3 = _ops.reduce_out3(out3)
                loss = out3[0].clone()

        saved = [zimg, t_prime, bias] + chunks
        idx = {}
        extra = []
        if save_g:
            if quant == "fp8":
                idx["g_chunks"] = len(extra)
                extra += g_chunks
                idx["out3s"] = len(extra)
                extra += out3s
                idx["gt_slab"] = len(extra)
                extra.append(gt_slab)
            else:
                idx["g_slab"] = len(extra)
                extra.append(g_slab)
                idx["out3"] = len(extra)
                extra.append(out3)
                if gt_slab is not None:
                    idx["gt_slab"] = len(extra)
                    extra.append(gt_slab)
        if use_q:
            # per-source (zt_q, st) + the shared (zi_q, si)
            idx["zi_q"] = len(extra)
            extra += [zi_q, si]
            idx["ztq"] = len(extra)
            for qc in qcaches:
                extra += [qc[2], qc[3]]
        ctx.save_for_backward(*saved, *extra)
        ctx.extra_idx = idx
        ctx.save_g = save_g
        ctx.group = group
        ctx.world = world
        ctx.rank = rank
        ctx.col_chunk = col_chunk
        ctx.impl = impl
        ctx.quant = quant
        ctx.b_txt = b_txt
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        from .. import ops as _ops
        world, rank = ctx.world, ctx.rank
        b = ctx.b_txt
        zimg, t_prime, bias = ctx.saved_tensors[:3]
        chunks = ctx.saved_tensors[3:3 + world]
        extra = ctx.saved_tensors[3 + world:]
        idx = ctx.extra_idx
        quant = ctx.quant

        # The cross-rank reduce-scatter (the collapse of the reference's
        # W−1 reversed ring hops, distributed_utils.py:74-77, into one RCCL
        # collective) is launched ASYNC from the on_dztxt hook, so its xGMI
        # wire time overlaps the dzimg gradient GEMM(s) still running on
        # the compute stream.
        comm = {}

        def on_dztxt(flat):
            if world == 1:
                return
            if _backend_is_gloo(ctx.group):
                comm["work"] = dist.all_reduce(
                    flat, op=dist.ReduceOp.SUM, group=ctx.group,
                    async_op=True)
                comm["flat"] = flat
            else:
                out = torch.empty_like(flat[:b])
                comm["work"] = dist.reduce_scatter_tensor(
                    out, flat, op=dist.ReduceOp.SUM, group=ctx.group,
                    async_op=True)
                comm["out"] = out

        go = grad_output
        if ctx.save_g and quant != "fp8":
            # One slab: dztxt_flat = t·go·(gᵀ @ zimg) first (feeds the async
            # reduce-scatter), then dzimg = t·go·(g @ concat-text).
            all_txt = chunks[0] if world == 1 else torch.cat(chunks, dim=0)
            out3 = extra[idx["out3"]]
            g = extra[idx["g_slab"]]
            gt = extra[idx["gt_slab"]] if "gt_slab" in idx else None
            qc = None
            if quant == "mixed":
                # mixed: g came from exact bf16 logits; quantization only
                # compresses the grad GEMM operands — one pass here.
                qc = _ops.quantize_fp8_pair(zimg, all_txt)
            dzimg, dtxt_flat, dt_prime, dbias = _ops.siglip_bwd_from_g(
                zimg, all_txt, t_prime, bias, go, out3, g, gt,
                quant=quant, qcache=qc, on_dztxt=on_dztxt)
        elif ctx.save_g:
            # fp8 saved-g: per-chunk g and scales (text side); the image
            # side has one scale, so dztxt is still a single GEMM over the
            # full gt slab.
            zi_q, si = extra[idx["zi_q"]], extra[idx["zi_q"] + 1]
            gt_slab = extra[idx["gt_slab"]]
            tp32 = t_prime.detach().reshape(()).float()
            t_true = tp32.exp()
            go32 = go.detach().reshape(()).to(zimg.device).float()
            scale = go32 * t_true
            dtxt_flat = _ops.scaled_mm8(gt_slab, zi_q, (scale / 448.0) * si)
            on_dztxt(dtxt_flat)
            dzimg_acc = torch.zeros_like(zimg, dtype=torch.float32)
            dt_acc = torch.zeros((), device=zimg.device, dtype=torch.float32)
            db_acc = torch.zeros((), device=zimg.device, dtype=torch.float32)
            for src in range(world):
                zt_q = extra[idx["ztq"] + 2 * src]
                st = extra[idx["ztq"] + 2 * src + 1]
                g_c = extra[idx["g_chunks"] + src]
                o3 = extra[idx["out3s"] + src]
                dzimg_acc += _ops.scaled_mm8(
                    g_c, zt_q, (scale / 448.0) * st).float()
                dt_acc = dt_acc + o3[1] * (t_true * si * st)
                db_acc = db_acc + o3[2]
            dzimg = dzimg_acc.to(zimg.dtype)
            dt_prime = (dt_acc * go32).to(t_prime.dtype).reshape(
                t_prime.shape)
            dbias = (db_acc * go32).to(bias.dtype).reshape(bias.shape)
        elif quant == "fp8" and "ztq" in idx:
            # fp8 recompute path: per-chunk backward with the forward's own
            # per-chunk scales (never requantize — fwd/bwd must see the same
            # quantized values).
            zi_q, si = extra[idx["zi_q"]], extra[idx["zi_q"] + 1]
            dzimg_acc = torch.zeros_like(zimg, dtype=torch.float32)
            dtxt_flat = torch.empty((world * b, zimg.shape[1]),
                                    device=zimg.device, dtype=zimg.dtype)
            dt_prime = None
            dbias = None
            for src in range(world):
                zt_q = extra[idx["ztq"] + 2 * src]
                st = extra[idx["ztq"] + 2 * src + 1]
                qc = (zi_q, si, zt_q, st)
                diag = 0 if src == rank else None
                dzi, dzt, dtp, dbs = chunk_loss_bwd(
                    zimg, chunks[src], t_prime, bias, diag, go,
                    col_chunk=ctx.col_chunk, impl=ctx.impl, quant=quant,
                    qcache=qc)
                dzimg_acc += dzi.float()
                dtxt_flat[src * b:(src + 1) * b] = dzt
                dt_prime = dtp if dt_prime is None else dt_prime + dtp
                dbias = dbs if dbias is None else dbias + dbs
            on_dztxt(dtxt_flat)
            dzimg = dzimg_acc.to(zimg.dtype)
        else:
            # bf16/mixed recompute: one fused backward over the concatenated
            # (W·b, d) text block — identical math to per-chunk calls (diag
            # at the own-rank block), a single dzimg accumulator and one
            # slab loop.
            all_txt = chunks[0] if world == 1 else torch.cat(chunks, dim=0)
            dzimg, dtxt_flat, dt_prime, dbias = chunk_loss_bwd(
                zimg, all_txt, t_prime, bias, rank * b, go,
                col_chunk=ctx.col_chunk, impl=ctx.impl, quant=quant,
                on_dztxt=on_dztxt)

        if world > 1:
            comm["work"].wait()
            if "out" in comm:
                dztxt = comm["out"]
            else:
                dztxt = comm["flat"][rank * b:(rank + 1) * b].clone()
        else:
            dztxt = dtxt_flat

        return (dzimg, dztxt, dt_prime, dbias, None, None, None, None, None,
                None)


class DistributedSigmoidLoss(nn.Module):
    """All-rank distributed sigmoid contrastive loss with module-owned params.

    Parity with reference ``DDPSigmoidLoss`` (``distributed_sigmoid_loss.py``):
    ``t_prime`` init ``log(10)``, ``bias`` init ``−10.0`` (``:11-12``); forward
    all-gathers the text shard, computes ``−logsigmoid(l·z)`` over the local
    ``(b, W·b)`` pair grid (labels ``+1`` on the rank-diagonal block's
    diagonal, ``−1`` elsewhere) and divides by the *local* ``gpu_batch_size``
    (``:47``).  Under DDP grad averaging this reproduces single-rank
    ``sum/(W·b)`` normalization exactly.

    ``strategy``:
        - ``"all_gather"`` — differentiable all-gather (one RCCL all-gather
          fwd, one reduce-scatter bwd), then ONE fused kernel call over the
          whole ``(b, W·b)`` block.
        - ``"ring"`` — pipelined P2P ring (xGMI point-to-point) with
          comm/compute overlap and reduce-scatter backward.

    Works with DistributedDataParallel (DDP) only (not DataParallel); pass
    ``t_prime``/``bias`` to your optimizer like any other parameter
    (reference ``README.md:19-20``).
    """

    def __init__(self, gpu_batch_size: int, strategy: str = "all_gather",
                 col_chunk: Optional[int] = None, impl: str = "auto",
                 quant: str = "bf16", bidir: Optional[bool] = None):
        super().__init__()
        self.t_prime = nn.Parameter(torch.tensor(math.log(10.0)))
        self.bias = nn.Parameter(torch.tensor(-10.0))
        self.gpu_batch_size = gpu_batch_size
        if strategy not in ("all_gather", "ring"):
            raise ValueError(f"unknown strategy {strategy!r}")
        if quant not in ("bf16", "fp8", "mixed"):
            raise ValueError(f"unknown quant {quant!r}")
        self.strategy = strategy
        self.col_chunk = col_chunk
        self.impl = impl
        self.quant = quant
        # Ring hop-halving via two simultaneous xGMI links (W > 2 only).
        # Default on; SIGLIP_RING_BIDIR=0 (or bidir=False) forces the
        # unidirectional ring for A/B comparison.
        if bidir is None:
            bidir = os.environ.get("SIGLIP_RING_BIDIR", "1") != "0"
        self.bidir = bidir
        if quant in ("fp8", "mixed") and torch.cuda.is_available():
            from .. import ops as _ops
            _ops.load_tuned_gemms()

    def forward(self, image_embeddings: torch.Tensor,
                text_embeddings: torch.Tensor, group=None) -> torch.Tensor:
        world, rank = _world_and_rank(group)
        b_txt = text_embeddings.shape[0]
        if self.strategy == "all_gather" or world == 1:
            all_txt = all_gather_with_grad(text_embeddings, group=group)
            total = sigmoid_contrastive_loss(
                image_embeddings, all_txt, self.t_prime, self.bias,
                diag_offset=rank * b_txt, col_chunk=self.col_chunk,
                impl=self.impl, quant=self.quant)
        else:
            want_grad = torch.is_grad_enabled() and any(
                t.requires_grad for t in (image_embeddings, text_embeddings,
                                          self.t_prime, self.bias))
            total = _RingAllGatherLoss.apply(
                image_embeddings, text_embeddings, self.t_prime, self.bias,
                group, self.col_chunk, self.impl, self.quant, want_grad,
                self.bidir)
        return total / self.gpu_batch_size


class SigLipLoss(nn.Module):
    """Ring-strategy SigLIP loss with caller-owned scale/bias parameters.

    Parity with the open_clip-style reference (``rwightman_sigmoid_loss.py``):
    local positive block first (``:69``), then ``W−1`` ring hops —
    bidirectional pairs plus one unidirectional remainder when ``bidir``
    (``:77-107``), else a pure unidirectional ring (``:108-122``) — each
    remote chunk contributing negatives-only loss, every chunk normalized by
    the local batch size (``:65``).  Gradients hop home through the
    autograd-reversed exchanges.

    ``use_horovod`` is unsupported, as in the reference (``:35``).
    """

    def __init__(self, cache_labels: bool = False, rank: int = 0,
                 world_size: int = 1, bidir: bool = True,
                 use_horovod: bool = False, impl: str = "auto",
                 col_chunk: Optional[int] = None):
        super().__init__()
        if use_horovod:
            raise NotImplementedError("horovod is not supported")
        self.cache_labels = cache_labels  # kept for API parity; predicate labels need no cache
        self.rank = rank
        self.world_size = world_size
        self.bidir = bidir
        self.use_horovod = use_horovod
        self.impl = impl
        self.col_chunk = col_chunk

    def get_ground_truth(self, device, dtype, num_logits,
                         negative_only=False) -> torch.Tensor:
        """Materialized ±1 label block — API parity with the reference
        (``rwightman_sigmoid_loss.py:43-47``).  The fused path never calls
        this (labels are an index predicate); it exists for users porting
        code that consumes it."""
        labels = -torch.ones((num_logits, num_logits), device=device,
                             dtype=dtype)
        if not negative_only:
            labels = 2 * torch.eye(num_logits, device=device,
                                   dtype=dtype) + labels
        return labels

    def get_logits(self, image_features, text_features, logit_scale,
                   logit_bias=None):
        """Pairwise logits — API parity with the reference
        (``rwightman_sigmoid_loss.py:49-53``)."""
        logits = logit_scale.exp() * image_features @ text_features.T
        if logit_bias is not None:
            logits = logits + logit_bias
        return logits

    def _loss(self, image_features, text_features, logit_scale, logit_bias,
              negative_only=False):
        off = None if negative_only else 0
        total = sigmoid_contrastive_loss(
            image_features, text_features, logit_scale, logit_bias,
            diag_offset=off, col_chunk=self.col_chunk, impl=self.impl)
        return total / image_features.shape[0]

    def forward(self, image_features, text_features, logit_scale, logit_bias,
                output_dict: bool = False):
        loss = self._loss(image_features, text_features, logit_scale,
                          logit_bias)

        if self.world_size > 1:
            right_rank = (self.rank + 1) % self.world_size
            left_rank = (self.rank - 1 + self.world_size) % self.world_size
            if self.bidir:
                to_right = to_left = text_features
                num_bidir, remainder = divmod(self.world_size - 1, 2)
                for _ in range(num_bidir):
                    recv = neighbour_exchange_bidir_with_grad(
                        left_rank, right_rank, to_left, to_right)
                    for f in recv:
                        loss = loss + self._loss(image_features, f,
                                                 logit_scale, logit_bias,
                                                 negative_only=True)
                    to_left, to_right = recv
                if remainder:
                    recv = neighbour_exchange_with_grad(
                        left_rank, right_rank, to_right)
                    loss = loss + self._loss(image_features, recv,
                                             logit_scale, logit_bias,
                                             negative_only=True)
            else:
                to_right = text_features
                for _ in range(self.world_size - 1):
                    from_left = neighbour_exchange_with_grad(
                        left_rank, right_rank, to_right)
                    loss = loss + self._loss(image_features, from_left,
                                             logit_scale, logit_bias,
                                             negative_only=True)
                    to_right = from_left

        return {"contrastive_loss": loss} if output_dict else loss
