"""RCCL collectives for DistEGNN: fused differentiable virtual-node exchange
and a flat gradient bucket.

Reference behavior being re-owned:
* ``_AllReduce`` differentiable autograd Function — forward all_reduce(SUM),
  backward all_reduce of the gradient (reference models/FastEGNN.py:10-43).
* ``weighted_average_reduce(data, weight)`` — node-count-weighted average of
  per-graph aggregates across ranks (reference models/FastEGNN.py:310-319),
  called 3 sites x n_layers per forward = 24 KB-scale all-reduces.
* DDP gradient averaging with the loss pre-scaled by ``world_size`` so the
  average becomes a sum (reference main.py:196, utils/train.py:110).

MI355X-first redesign: xGMI is point-to-point (7 links/GPU) and these
payloads are KBs, so they are latency-bound — the win is FEWER collectives,
not bigger ones. We therefore
(1) fuse each layer's virtual-state tensors into ONE flat fp32 buffer per
    all-reduce site (``fused_weighted_average_reduce``), cutting 6
    all-reduces per layer (data+weight x 3 sites) to 2;
(2) compute per-graph node counts ONCE per step (``global_counts``) instead
    of re-reducing the weights at every site (they are step-constants);
(3) replace DDP with a single flat-bucket gradient all_reduce(SUM) at the
    optimizer-step boundary (``GradBucket``) — identical math to the
    reference's average x world_size pre-scaling, without per-backward
    bucket traffic or ``find_unused_parameters`` overhead.
"""

from __future__ import annotations

import datetime
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None,
                     timeout_s: int = 18000) -> tuple[int, int]:
    """Initialize the process group from torchrun env vars.

    Returns (local_rank, world_size). Single-process (no env) → (0, 1)
    without creating a group. Backend defaults to nccl (=RCCL on ROCm) when
    CUDA devices are visible, else gloo.
    """
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    force = os.environ.get("DISTEGNN_FORCE_DIST") == "1"
    if world_size <= 1 and not (force and "RANK" in os.environ):
        return 0, 1
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        dist.init_process_group(
            backend,
            timeout=datetime.timedelta(seconds=timeout_s),
        )
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
        global _GRAPH_PG
        if _GRAPH_PG is None:
            # second communicator reserved for captured collectives (see
            # _capture_group); warm it eagerly — NCCL comm init during a
            # stream capture would be fatal
            _GRAPH_PG = dist.new_group(backend="nccl")
            warm = torch.ones(1, device=f"cuda:{local_rank}")
            dist.all_reduce(warm, group=_GRAPH_PG)
            torch.cuda.synchronize()
    return local_rank, world_size


_GRAPH_PG = None  # dedicated communicator for hipGraph-captured collectives


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


_FORCE_CAPTURE_COMM = False


def _capture_group():
    """The communicator for the CURRENT collective call.

    NCCL/RCCL forbids mixing graph-captured and eager collectives on one
    communicator. Calls issued while a stream capture is active use a
    dedicated process group (created and warmed eagerly at init); the
    same group is used under ``capture_comm_fallback`` so that a rank
    whose capture failed keeps issuing the SAME collective sequence on
    the SAME communicator as the ranks that replay captured graphs
    (asymmetric fallback must not become a comm-mismatch deadlock)."""
    if _GRAPH_PG is not None and (
            _FORCE_CAPTURE_COMM
            or (torch.cuda.is_available()
                and torch.cuda.is_current_stream_capturing())):
        return _GRAPH_PG
    return None  # default group


class capture_comm_fallback:
    """Context manager: route collectives to the capture communicator
    even though no capture is active (eager fallback of a graphed step)."""

    def __enter__(self):
        global _FORCE_CAPTURE_COMM
        self._prev = _FORCE_CAPTURE_COMM
        _FORCE_CAPTURE_COMM = True
        return self

    def __exit__(self, *exc):
        global _FORCE_CAPTURE_COMM
        _FORCE_CAPTURE_COMM = self._prev
        return False


def world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def barrier():
    if is_distributed():
        dist.barrier()


def destroy():
    global _GRAPH_PG
    if is_distributed():
        dist.destroy_process_group()
    _GRAPH_PG = None


class _FusedAllReduceSum(torch.autograd.Function):
    """Differentiable SUM all-reduce of several tensors through ONE flat
    buffer (one RCCL call forward, one mirrored call backward).

    Equivalent math to the reference's per-tensor ``_AllReduce``
    (models/FastEGNN.py:10-21) applied to each input, but latency-optimal
    for xGMI's point-to-point links."""

    @staticmethod
    def forward(ctx, *tensors: torch.Tensor):
        if not is_distributed():
            return tensors if len(tensors) > 1 else tensors[0]
        flat = torch.cat([t.reshape(-1) for t in tensors])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=_capture_group())
        outs = []
        ofs = 0
        for t in tensors:
            n = t.numel()
            outs.append(flat[ofs:ofs + n].view_as(t))
            ofs += n
        return tuple(outs) if len(outs) > 1 else outs[0]

    @staticmethod
    def backward(ctx, *grads: torch.Tensor):
        if not is_distributed():
            return grads if len(grads) > 1 else grads[0]
        flat = torch.cat([g.reshape(-1) for g in grads])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=_capture_group())
        outs = []
        ofs = 0
        for g in grads:
            n = g.numel()
            outs.append(flat[ofs:ofs + n].view_as(g))
            ofs += n
        return tuple(outs)


def _drain_watchdog():
    """Let the ProcessGroupNCCL watchdog dequeue outstanding works before a
    capture (its hipEventQuery during capture aborts the process).

    LOCAL drain only — deliberately NO dist.barrier(): the watchdog
    watches this rank's works, which a device synchronize completes in a
    lockstep schedule (peers have issued the matching collectives by
    construction). A default-group barrier here can deadlock when ranks
    reach their capture points at different step indices (per-rank shape
    keys: one rank's batches may collide in shape and hit the
    warmup-occurrence threshold earlier than its peers, who are busy
    issuing capture-communicator collectives and never touch the default
    group inside the replay window)."""
    import time as _time

    torch.cuda.synchronize()
    _time.sleep(0.5)


def all_reduce_sum_differentiable(*tensors: torch.Tensor):
    """Fused differentiable all_reduce(SUM). Identity when world_size == 1."""
    return _FusedAllReduceSum.apply(*tensors)


def global_counts(counts: torch.Tensor) -> torch.Tensor:
    """All-reduce the per-graph node counts [B] once per step.

    The reference recomputes these weights with a host-synced python loop at
    every reduce site (FastEGNN.py:196,226,260); they are constant within a
    step, so one tiny collective replaces 12+ host syncs + 24 reduces.

    Routed through ``_capture_group()``: when issued inside a hipGraph
    capture (bench/trainer step_core) or under ``capture_comm_fallback``,
    the reduce lands on the dedicated capture communicator — mixing it onto
    the default group would violate NCCL's captured/eager single-communicator
    rule (the teardown barrier and timing reduces run eagerly on the default
    group)."""
    if not is_distributed():
        return counts
    out = counts.clone()
    dist.all_reduce(out, op=dist.ReduceOp.SUM, group=_capture_group())
    return out


class CapturedAllReduce:
    """All-reduce as a captured-graph replay, for collectives that must run
    INSIDE the replay window of an RCCL-containing step graph (per-epoch
    logging reduce, early-stop flag — profiles/README.md roadmap #2).

    An eager collective on the DEFAULT communicator between replays of a
    captured RCCL graph corrupts replayed outputs on this stack (bisected:
    tools/fd_debug.py --barrier). Replaying a pre-captured graph on the
    capture communicator keeps every post-capture collective on one
    communicator with a rank-consistent schedule.

    ``prebuild`` MUST be called before the first step-graph capture: building
    lazily would interleave the build's watchdog drain (a default-group
    barrier) into the replay window. Non-CUDA / non-distributed calls fall
    back to an eager all_reduce (gloo path, CPU tests)."""

    def __init__(self):
        self._entries = {}

    @staticmethod
    def _key(t: torch.Tensor, op) -> tuple:
        return (tuple(t.shape), t.dtype, str(op))

    def _build(self, t: torch.Tensor, op):
        buf = t.clone()
        _drain_watchdog()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            dist.all_reduce(buf, op=op, group=_capture_group())
        self._entries[self._key(t, op)] = (g, buf)

    def prebuild(self, tensors_ops):
        """tensors_ops: iterable of (example_tensor, ReduceOp). Call during
        setup, before any step-graph capture."""
        if not is_distributed() or not torch.cuda.is_available():
            return
        for t, op in tensors_ops:
            if t.is_cuda and self._key(t, op) not in self._entries:
                self._build(t, op)

    def rebuild(self):
        """Drop + rebuild every captured all-reduce graph. Call after any
        device-wide sync between replays (eval epochs, checkpoint D2H):
        a hipDeviceSynchronize garbles existing graph execs on this stack
        (runtime/graphs.GraphedStep.invalidate). Rebuild EAGERLY (not
        lazily) so the builds' own pre-capture syncs land before any step
        graph recaptures."""
        if not self._entries:
            return
        specs = [(buf, key[2]) for key, (g, buf) in self._entries.items()]
        self._entries = {}
        for buf, op_str in specs:
            op = next((o for o in (dist.ReduceOp.SUM, dist.ReduceOp.MAX,
                                   dist.ReduceOp.MIN, dist.ReduceOp.AVG,
                                   dist.ReduceOp.PRODUCT)
                       if str(o) == op_str), None)
            if op is None:
                # dropping an entry silently would desync the per-rank
                # collective schedule on the next replay
                raise RuntimeError(
                    f"CapturedAllReduce.rebuild: unknown op {op_str!r}")
            self._build(buf, op)

    def __call__(self, t: torch.Tensor, op=None) -> torch.Tensor:
        op = op if op is not None else dist.ReduceOp.SUM
        if not is_distributed():
            return t
        if not (torch.cuda.is_available() and t.is_cuda):
            out = t.clone()
            dist.all_reduce(out, op=op)
            return out
        entry = self._entries.get(self._key(t, op))
        if entry is None:  # late build: caller did not prebuild
            self._build(t, op)
            entry = self._entries[self._key(t, op)]
        g, buf = entry
        buf.copy_(t)
        g.replay()
        # the STATIC buffer is returned (no allocation inside the replay
        # window): read/copy it before the next call with the same key
        return buf


def fused_weighted_average_reduce(tensors: Sequence[torch.Tensor],
                                  counts: torch.Tensor,
                                  counts_sum: torch.Tensor):
    """Node-count-weighted average of per-graph aggregates across ranks.

    Each tensor is [B, ...] (per-graph local MEANS over this rank's
    partition nodes). Returns the global per-graph means:
        out = all_reduce_sum(t * counts) / counts_sum
    Gradient flows through the differentiable sum (weights are constants),
    matching the composite gradient of the reference's
    ``weighted_average_reduce`` (FastEGNN.py:310-319): g_in = w/W * AR(g_out).
    """
    if not is_distributed():
        return tuple(tensors) if len(tensors) > 1 else tensors[0]
    b = counts.shape[0]
    scaled = []
    for t in tensors:
        w = counts.view((b,) + (1,) * (t.dim() - 1)).to(t.dtype)
        scaled.append(t * w)
    reduced = _FusedAllReduceSum.apply(*scaled)
    if len(tensors) == 1:
        reduced = (reduced,)
    outs = []
    for t in reduced:
        w = counts_sum.view((b,) + (1,) * (t.dim() - 1)).to(t.dtype)
        outs.append(t / w.clamp(min=1))
    return tuple(outs) if len(outs) > 1 else outs[0]


class GradBucket:
    """Flat single-bucket gradient all-reduce (replaces DDP for these models).

    The reference wraps the model in DDP (find_unused_parameters=True) which
    AVERAGES gradients every backward, and multiplies the loss by world_size
    to recover SUM semantics (utils/train.py:110). Model size here is ~0.5 MB
    (H=64), so one flat all_reduce(SUM) per optimizer step is both exact and
    minimal. Call ``sync()`` after the last micro-batch backward, before
    clipping/stepping.
    """

    def __init__(self, model: torch.nn.Module):
        self.params: List[torch.nn.Parameter] = [
            p for p in model.parameters() if p.requires_grad
        ]

    def broadcast_parameters(self):
        """Rank-0 parameter broadcast at startup (DDP does this implicitly)."""
        if not is_distributed():
            return
        flat = torch.cat([p.data.reshape(-1) for p in self.params])
        dist.broadcast(flat, src=0)
        ofs = 0
        for p in self.params:
            n = p.numel()
            p.data.copy_(flat[ofs:ofs + n].view_as(p))
            ofs += n

    def prebuild_graph_sync(self):
        """Build the captured grad-sync graph during setup, BEFORE the first
        step-graph capture — a lazy build's watchdog drain (default-group
        barrier) must not land inside the replay window."""
        if is_distributed() and torch.cuda.is_available():
            self.graph_sync(_build_only=True)

    def rebuild_graph_sync(self):
        """Drop + eagerly rebuild the captured grad-sync graph after a
        device-wide sync between replays (see CapturedAllReduce.rebuild)."""
        if getattr(self, "_sync_graph", None) is not None:
            self._sync_graph = None
            self.prebuild_graph_sync()

    @torch.no_grad()
    def graph_sync(self, _build_only: bool = False):
        """sync() as a captured graph replay (prebuild at setup via
        ``prebuild_graph_sync``; lazily built otherwise). Falls back to
        eager sync when not distributed or not on CUDA."""
        if not is_distributed():
            return
        if not torch.cuda.is_available():
            return self.sync()
        if getattr(self, "_sync_graph", None) is None:
            for p in self.params:
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
            torch.cuda.synchronize()
            _drain_watchdog()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                flat = torch.cat([p.grad.reshape(-1) for p in self.params])
                dist.all_reduce(flat, op=dist.ReduceOp.SUM,
                                group=_capture_group())
                ofs = 0
                for p in self.params:
                    n = p.numel()
                    p.grad.copy_(flat[ofs:ofs + n].view_as(p.grad))
                    ofs += n
            self._sync_graph = g
        if not _build_only:
            self._sync_graph.replay()

    def sync(self):
        if not is_distributed():
            return
        grads = []
        for p in self.params:
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            grads.append(p.grad.reshape(-1))
        flat = torch.cat(grads)
        dist.all_reduce(flat, op=dist.ReduceOp.SUM)
        ofs = 0
        for p in self.params:
            n = p.numel()
            p.grad.copy_(flat[ofs:ofs + n].view_as(p))
            ofs += n


def check_model_parameters(model: torch.nn.Module) -> bool:
    """Cross-rank parameter consistency check (reference main.py:40-55)."""
    if not is_distributed():
        return True
    params = torch.cat([p.data.reshape(-1) for p in model.parameters()])
    ref = params.clone()
    dist.broadcast(ref, src=0)
    return bool(torch.allclose(params, ref, atol=1e-6))
