"""Low-precision all-reduce with custom-precision partial sums.

The reference *emulates* a low-precision all-reduce by all-gathering every
rank's full gradient and replaying a sequential quantized summation on every
rank (dist_util.py:60-89): (W-1)x the wire traffic of a ring all-reduce plus
W full-tensor quantize passes per rank.

Here the reduction is REAL: a ring reduce-scatter over RCCL/xGMI where each
hop applies the quantized accumulate Q(acc + g) (or the quantized Kahan
triple) with a HIP kernel, followed by an all-gather of the reduced chunks.
xGMI links are point-to-point (~153 GB/s per direction per GPU), so the ring
moves 2(W-1)/W ~= 2x the bucket per rank instead of the emulation's (W-1)x,
and with the bf16 wire format (exact for man_bits <= 7 — every (exp<=8,m<=7)
grid value is a bf16) halves that again.

Semantics contract (documented divergence, SURVEY.md §7 hard-part 2):
  * mode="sequential": bit-identical to the reference emulation — all-gather
    + sequential quantized sum in rank order 0..W-1 on every rank.
  * mode="ring": chunk i's summation starts at rank i and proceeds in ring
    order (i, i+1, ..., i-1): the same W quantized accumulation steps as the
    sequential mode, in a rotated rank order per chunk.  Under APS the
    addends share one power-of-two pre-scale, so the rounding behavior is
    the same class; bit-identity with the sequential order is only
    guaranteed per-chunk up to that rotation (validated by tests against
    the rotated sequential oracle).
"""
import os

import torch
import torch.distributed as dist

from .. import ops


def _wire_dtype(flat, man_bits, wire):
    # bf16 wire is only EXACT when the contributions are already on an
    # (exp<=8, man<=7) grid (e.g. after the APS scale+quantize pass), so it is
    # opt-in: callers that know the grads are on-grid pass wire="bf16"
    # (sum_gradients does).  Default is the always-exact f32 wire.
    if wire == "bf16" and man_bits > 7:
        raise ValueError(
            f"wire='bf16' cannot represent man={man_bits} > 7 grids exactly; "
            "use the f32 wire")
    if wire is not None:
        return {"bf16": torch.bfloat16, "f32": torch.float32}[wire]
    return torch.float32


def _check_on_grid(flat, wdt):
    # CPD_DEBUG_WIRE=1: verify the caller's on-grid claim — the bf16
    # downcast must round-trip exactly (one pass over the bucket; debug only)
    if wdt is torch.bfloat16 and os.environ.get("CPD_DEBUG_WIRE") == "1":
        if not torch.equal(flat.to(wdt).float(), flat):
            raise RuntimeError(
                "wire='bf16' requested but data is not on a bf16 grid "
                "(caller must APS-scale+quantize to man<=7 first)")


def _qadd(acc, inc, man, exp):
    if acc.dtype == torch.bfloat16:
        ops.ext_for(acc).qadd_bf16_(acc, inc, man, exp)
    else:
        ops.qadd_(acc, inc, man, exp)


def _kahan_qadd(acc, comp, inc, man, exp):
    if acc.dtype == torch.bfloat16:
        ops.ext_for(acc).kahan_qadd_bf16_(acc, comp, inc, man, exp)
    else:
        ops.kahan_qadd_(acc, comp, inc, man, exp)


def _quantize_chunk_(chunk, man, exp):
    """chunk = Q(chunk) in the chunk's wire dtype."""
    if chunk.dtype == torch.bfloat16:
        # Q(0 + x) through the bf16 hop kernel == Q(x)
        zero = torch.zeros_like(chunk)
        ops.ext_for(chunk).qadd_bf16_(chunk, zero, man, exp)
    else:
        ops.quantize_(chunk, man, exp)


def sequential_lp_all_reduce_(flat, grad_exp, grad_man, use_kahan=False,
                              group=None):
    """Reference-emulation semantics: all-gather + in-order quantized sum
    (dist_util.py:60-89).  Bit-identical on every rank by construction."""
    if dist.is_available() and dist.is_initialized():
        W = dist.get_world_size(group)
        gather = [torch.empty_like(flat) for _ in range(W)]
        dist.all_gather(gather, flat, group=group)
    else:
        gather = [flat.clone()]
    res = torch.zeros_like(flat)
    if use_kahan:
        comp = torch.zeros_like(flat)
        for g in gather:
            ops.kahan_qadd_(res, comp, g, grad_man, grad_exp)
    else:
        for g in gather:
            ops.qadd_(res, g, grad_man, grad_exp)
    flat.copy_(res)
    return flat


def ring_lp_all_reduce_(flat, grad_exp, grad_man, use_kahan=False, group=None,
                        wire=None):
    """Real ring all-reduce with per-hop quantized partial sums."""
    if dist.is_available() and dist.is_initialized():
        W = dist.get_world_size(group)
        r = dist.get_rank(group)
    else:
        W, r = 1, 0
    man, exp = grad_man, grad_exp
    n = flat.numel()

    wdt = _wire_dtype(flat, man, wire)
    _check_on_grid(flat, wdt)
    chunk = (n + W - 1) // W
    padded = chunk * W
    if wdt == torch.float32 and padded == n:
        work = flat
    else:
        work = torch.zeros(padded, dtype=wdt, device=flat.device)
        work[:n].copy_(flat)  # exact downcast for on-grid values when bf16
    chunks = work.view(W, chunk)

    if W == 1:
        _quantize_chunk_(chunks[0], man, exp)
        if work is not flat:
            flat.copy_(work[:n])
        return flat

    right = (r + 1) % W
    left = (r - 1) % W

    if not use_kahan:
        # chunk i originates at rank i: one entry quantize, then W-1 hops of
        # Q(partial + local) accumulated in place.
        _quantize_chunk_(chunks[r], man, exp)
        recv = torch.empty(chunk, dtype=wdt, device=flat.device)
        for s in range(W - 1):
            send_idx = (r - s) % W
            recv_idx = (r - s - 1) % W
            reqs = dist.batch_isend_irecv([
                dist.P2POp(dist.isend, chunks[send_idx], right, group=group),
                dist.P2POp(dist.irecv, recv, left, group=group),
            ])
            for q in reqs:
                q.wait()
            _qadd(chunks[recv_idx], recv, man, exp)
    else:
        # Kahan: the (acc, comp) pair travels the ring (2x wire in this
        # phase); each hop folds the local raw chunk in with the quantized
        # Kahan triple.  Same W steps as the sequential oracle.
        trav = torch.zeros(2, chunk, dtype=wdt, device=flat.device)
        _kahan_qadd(trav[0], trav[1], chunks[r], man, exp)
        recv = torch.empty(2, chunk, dtype=wdt, device=flat.device)
        for s in range(W - 1):
            reqs = dist.batch_isend_irecv([
                dist.P2POp(dist.isend, trav, right, group=group),
                dist.P2POp(dist.irecv, recv, left, group=group),
            ])
            for q in reqs:
                q.wait()
            local_idx = (r - s - 1) % W
            _kahan_qadd(recv[0], recv[1], chunks[local_idx], man, exp)
            trav, recv = recv, trav
        chunks[(r + 1) % W].copy_(trav[0])

    # all-gather the reduced chunks; rank q owns final chunk (q+1) % W, so
    # roll the output list to land each chunk at its home position.
    out = [chunks[(q + 1) % W] for q in range(W)]
    dist.all_gather(out, chunks[(r + 1) % W].clone(), group=group)

    if work is not flat:
        flat.copy_(work[:n])
    return flat


def lp_all_reduce_(flat, grad_exp, grad_man, use_kahan=False, mode="ring",
                   group=None, wire=None):
    if mode == "sequential":
        return sequential_lp_all_reduce_(flat, grad_exp, grad_man,
                                         use_kahan=use_kahan, group=group)
    elif mode == "ring":
        return ring_lp_all_reduce_(flat, grad_exp, grad_man,
                                   use_kahan=use_kahan, group=group, wire=wire)
    raise ValueError(f"unknown lp_all_reduce mode {mode!r}")
