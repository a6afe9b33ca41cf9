"""Pipeline point-to-point communication.

Reference: galvatron/core/runtime/pipeline/pipeline.py:1091-1591
(_run_p2pops, fused send-recv).  On the MI355X node each pp hop is one xGMI
link; batch_isend_irecv maps to RCCL p2p.  Shapes are computed analytically
from the plan on both sides (no dynamic negotiation needed; the reference's
shape handshake exists because its runtime lacks a global plan view at p2p
time).
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


def _p2p(ops: List[dist.P2POp]) -> None:
    if not ops:
        return
    reqs = dist.batch_isend_irecv(ops)
    for r in reqs:
        r.wait()


def send_tensor(t: torch.Tensor, dst: int) -> None:
    _p2p([dist.P2POp(dist.isend, t.contiguous(), dst)])


def send_tensor_async(t: torch.Tensor, dst: int):
    """Non-blocking send: returns (reqs, tensor) — the caller must keep the
    tensor alive and drain the reqs before the step ends (1F1B warmup/
    cooldown sends overlap the next microbatch's compute)."""
    tc = t.contiguous()
    reqs = dist.batch_isend_irecv([dist.P2POp(dist.isend, tc, dst)])
    return reqs, tc


def send_recv_async(send: Optional[torch.Tensor], send_to: Optional[int],
                    recv_shape, recv_dtype, recv_from: Optional[int], device):
    """One batch, but only the recv is waited; the send's reqs are returned
    for deferred draining (reference fused send-fwd-recv-bwd, with the
    send leg made non-blocking)."""
    ops = []
    buf = None
    tc = None
    if send is not None and send_to is not None:
        tc = send.contiguous()
        ops.append(dist.P2POp(dist.isend, tc, send_to))
    if recv_shape is not None and recv_from is not None:
        buf = torch.empty(*recv_shape, dtype=recv_dtype, device=device)
        ops.append(dist.P2POp(dist.irecv, buf, recv_from))
    if not ops:
        return buf, [], None
    reqs = dist.batch_isend_irecv(ops)
    send_reqs = []
    if buf is not None:
        reqs[-1].wait()
        send_reqs = reqs[:-1]
    else:
        send_reqs = reqs
    return buf, send_reqs, tc


def recv_tensor(shape, dtype, src: int, device) -> torch.Tensor:
    buf = torch.empty(*shape, dtype=dtype, device=device)
    _p2p([dist.P2POp(dist.irecv, buf, src)])
    return buf


def send_recv(send: Optional[torch.Tensor], send_to: Optional[int],
              recv_shape, recv_dtype, recv_from: Optional[int], device
              ) -> Optional[torch.Tensor]:
    """Fused send+recv in one batch (1F1B steady state: send fwd / recv bwd
    etc.; reference: pipeline.py:1350-1591)."""
    ops = []
    buf = None
    if send is not None and send_to is not None:
        ops.append(dist.P2POp(dist.isend, send.contiguous(), send_to))
    if recv_shape is not None and recv_from is not None:
        buf = torch.empty(*recv_shape, dtype=recv_dtype, device=device)
        ops.append(dist.P2POp(dist.irecv, buf, recv_from))
    _p2p(ops)
    return buf
