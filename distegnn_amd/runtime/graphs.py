"""hipGraph-captured training step (HIP graphs instead of a tracing compiler).

The training step launches ~1300 kernels; at ~8-15 us host cost per launch
the CPU becomes the bottleneck once the kernels are fast (measured ~22 ms
host gap vs 37 ms GPU time on the LargeFluid workload). ``GraphedStep``
captures the forward+loss+backward of one mini-batch shape into a hipGraph
and replays it with one launch.

Design:
* Shape-keyed cache: geometric graphs vary in node/edge count across
  samples, so a graph is captured per (num_nodes, num_edges, num_graphs)
  key after the same shape has been seen ``warmup_occurrences`` times
  (allocator warm, hipBLASLt workspaces settled). Unseen shapes run eagerly.
* Static buffers: the batch's tensors are copied into capture-owned
  buffers before replay (device-to-device, async).
* Gradient accumulation composes naturally: `.grad` buffers are
  pre-materialized and the captured backward accumulates into them
  (`zero_grad(set_to_none=False)` keeps the addresses stable). The
  optimizer step runs OUTSIDE the graph (it executes every
  accumulation_steps only, on 0.5 MB of parameters).
* RNG: random sampling runs OUTSIDE the captured region — callers draw
  per-step randomness eagerly (losses.draw_sample_indices) and pass it as
  a static input (mmd_idx / mmd_valid batch fields). In-graph RNG would
  replay a frozen philox offset, repeating the same "random" sample every
  step.
* Eager work BETWEEN replays (optimizer step, grad clip) must go through
  ``run_eager``: on this ROCm stack the allocator was observed to hand
  default-stream blocks to eager ops while a captured graph's private
  pool still used them (Adam temporaries aliased graph blocks after ~11
  replays — loss scaled wrong, params eventually NaN). ``run_eager``
  runs the callable on a dedicated side stream whose per-stream blocks
  never overlap the graph pool.
* Distributed: the virtual-node RCCL all-reduces are captured INSIDE the
  graph (validated on a 1-rank group via bench.py DISTEGNN_FORCE_DIST=1).
  Asymmetric capture failure cannot deadlock ranks: the eager fallback
  re-issues the same collectives in the same program order.

Limitations: replay requires the SAME dtypes/fields per shape key; any
capture failure disables capture for that key and logs once.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

_BATCH_FIELDS = ("x", "pos", "vel", "attr", "target", "edge_index",
                 "edge_attr", "batch", "ptr", "rowptr", "colptr", "col_perm",
                 "counts", "counts_global", "loc_mean", "pool_chunk_begin",
                 "pool_chunk_end", "pool_seg_chunk_ptr", "mmd_idx",
                 "mmd_valid")


class _ShapeEntry:
    def __init__(self):
        self.seen = 0
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.static = None       # dict of static input buffers
        self.outputs = None      # eager-owned output buffers
        self.out_meta = None     # [(shape, dtype), ...] from last warmup
        self.disabled = False


class GraphedStep:
    """Capture/replay wrapper around a step_fn(batch_buffers) -> (loss, aux).

    step_fn must: read ONLY the tensors handed to it, run forward + loss +
    backward() internally, and return (loss, aux) tensors.
    """

    def __init__(self, step_fn, params, warmup_occurrences: int = 2,
                 enabled: bool = True, verbose: bool = False,
                 fallback_ctx=None, max_entries: int = 32):
        self.step_fn = step_fn
        self.params = list(params)
        self.warmup = warmup_occurrences
        self.enabled = enabled and torch.cuda.is_available()
        self.verbose = verbose
        self.entries: Dict[Tuple, _ShapeEntry] = {}
        self._side = None  # side stream for eager work between replays
        # context manager factory wrapped around EVERY eager invocation
        # while capture is enabled (warmup occurrences, post-capture
        # fallbacks, overflow keys): per-rank shape keys mean one rank can
        # be replaying (collectives on the capture communicator) while
        # another runs the same step eagerly — routing the eager rank's
        # collectives to the same communicator keeps the per-communicator
        # schedule rank-consistent (comm mismatch would deadlock)
        self.fallback_ctx = fallback_ctx
        # graph pools cannot be freed safely; real datasets have per-sample
        # shapes, so cap captured entries — overflow keys run eagerly (with
        # fallback_ctx, so the collective schedule stays consistent)
        self.max_entries = max_entries
        self._overflow_warned = False

    def run_eager(self, fn):
        """Run eager work (optimizer step, clip, zero_grad) between replays.

        MUST be used for any eager op that allocates device memory between
        replays of a captured graph. On this ROCm stack the caching
        allocator hands default-stream blocks to eager ops even though a
        captured graph's private pool still references them: Adam's
        foreach temporaries started aliasing graph blocks after ~11
        replays, silently scaling the loss (and eventually NaN-ing
        parameters). Running the eager work on a dedicated side stream
        keeps its allocations in per-stream blocks that never overlap the
        graph pool (verified: 30 replays bit-exact vs eager,
        tools/graph_bisect3.py adam_side)."""
        if not self.enabled:
            return fn()
        if self._side is None:
            self._side = torch.cuda.Stream()
            # Dedicated allocator pool for eager work between replays.
            # The side stream alone is NOT sufficient isolation: a large
            # eager region (an fp32 eval epoch at 113K nodes allocates
            # hundreds of MB) was bisected to corrupt replayed training
            # state on this stack even on the side stream — the fourth
            # replay-window hazard. Allocations drawn from a private
            # MemPool can never alias blocks the captured graphs
            # reference. (Fallback: side stream only, as in round 1.)
            try:
                self._eager_pool = torch.cuda.MemPool()
            except Exception:
                self._eager_pool = None
        self._side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._side):
            if self._eager_pool is not None:
                with torch.cuda.memory.use_mem_pool(self._eager_pool):
                    out = fn()
            else:
                out = fn()
        torch.cuda.current_stream().wait_stream(self._side)
        return out

    def invalidate(self, reason: str = ""):
        """Drop every captured graph; each shape recaptures on its next
        occurrence (no extra eager warmups — out_meta is kept).

        MUST be called after any operation that device-synchronizes
        between replays. On this ROCm stack a hipDeviceSynchronize
        garbles existing graph execs: a bare torch.cuda.synchronize()
        (or a pageable D2H read, e.g. torch.save / .cpu() on params)
        between replays NaNs the training trajectory about one epoch
        later even though the very next epoch of replays is still
        bit-exact (bisected via DISTEGNN_DBG_EVAL_MODE=sync/d2h;
        profiles/README.md replay-window hazard #3). Eval epochs and
        checkpoint saves therefore invalidate + recapture."""
        n = 0
        for e in self.entries.values():
            if e.graph is not None:
                e.graph = None
                e.static = None
                e.outputs = None
                n += 1
        if n and self.verbose:
            print(f"[graphs] invalidated {n} captured graph(s)"
                  f"{': ' + reason if reason else ''}")
        return n

    def _fallback(self, batch):
        if self.fallback_ctx is None:
            return self.step_fn(batch)
        with self.fallback_ctx():
            return self.step_fn(batch)

    @staticmethod
    def _key(batch) -> Tuple:
        # (num_nodes, num_edges, num_graphs) alone does not distinguish
        # batches whose optional static fields differ in shape or presence
        # (e.g. pool-chunk tables at equal totals) — _copy_into would then
        # copy_ with mismatched shapes at replay. Key on the full field
        # signature instead.
        sig = tuple(
            (f, tuple(v.shape), v.dtype)
            for f in _BATCH_FIELDS
            if torch.is_tensor(v := getattr(batch, f, None)))
        return (batch.num_nodes, batch.num_edges, batch.num_graphs, sig)

    def _snapshot(self, batch) -> dict:
        out = {}
        for f in _BATCH_FIELDS:
            v = getattr(batch, f, None)
            if torch.is_tensor(v):
                out[f] = v.clone()
        return out

    def _copy_into(self, static: dict, batch):
        for f, buf in static.items():
            buf.copy_(getattr(batch, f), non_blocking=True)

    def __call__(self, batch):
        if not self.enabled:
            return self.step_fn(batch)
        # captured graphs read version-cached weight transforms
        # (ops/prep.py); if the optimizer stepped since the last refresh,
        # refresh them on the side stream BEFORE replaying — replaying
        # against stale weights would silently train wrong. Host-only
        # version scan when nothing changed.
        from ..ops import prep as _prep

        if _prep.any_stale():
            self.run_eager(_prep.refresh)
        key = self._key(batch)
        e = self.entries.get(key)
        if e is None:
            captured = sum(1 for v in self.entries.values()
                           if v.graph is not None or not v.disabled)
            if captured >= self.max_entries:
                if not self._overflow_warned:
                    self._overflow_warned = True
                    print(f"[graphs] shape-key cache full "
                          f"({self.max_entries}); further shapes run eager")
                return self._fallback(batch)
            e = self.entries.setdefault(key, _ShapeEntry())
        if e.disabled:
            return self._fallback(batch)
        if e.graph is not None:
            self._copy_into(e.static, batch)
            e.graph.replay()
            return e.outputs
        e.seen += 1
        if e.seen <= self.warmup or e.out_meta is None:
            # warmup occurrences run eagerly; collectives still go to the
            # capture communicator (another rank may already be replaying)
            outs = self._fallback(batch)
            e.out_meta = [(tuple(o.shape), o.dtype, o.device)
                          for o in outs]
            return outs
        # capture attempt
        try:
            static = self._snapshot(batch)

            class _Proxy:
                pass

            proxy = _Proxy()
            for f, v in static.items():
                setattr(proxy, f, v)
            for f in ("num_graphs",):
                setattr(proxy, f, getattr(batch, f))
            proxy.num_nodes = batch.num_nodes
            proxy.num_edges = batch.num_edges
            # ensure grad buffers exist before capture
            for p in self.params:
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
            # Output buffers are allocated EAGERLY (normal allocator pool)
            # before capture and the captured region copies into them —
            # exactly like the pre-materialized .grad buffers. Returning
            # tensors that live in the graph's private mempool is unsafe on
            # this ROCm stack: post-replay eager work (Adam temporaries)
            # was observed to scribble private-pool blocks, corrupting the
            # logged loss while training itself stayed correct.
            out_bufs = tuple(torch.empty(shape, dtype=dtype, device=dev)
                             for (shape, dtype, dev) in e.out_meta)
            torch.cuda.synchronize()
            if (torch.distributed.is_available()
                    and torch.distributed.is_initialized()):
                # Drain the ProcessGroupNCCL watchdog before capturing:
                # the watchdog thread polls hipEventQuery on outstanding
                # pre-capture collectives and an event query DURING stream
                # capture is hipErrorStreamCaptureUnsupported -> process
                # abort. Collectives issued INSIDE the capture are not
                # watched. LOCAL drain only — a default-group barrier here
                # deadlocks when shape-key collisions make ranks reach
                # their capture points at different step indices (see
                # parallel/comm._drain_watchdog).
                import time as _time

                torch.cuda.synchronize()
                _time.sleep(0.5)
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                outputs = self.step_fn(proxy)
                for buf, o in zip(out_bufs, outputs):
                    buf.copy_(o)
            e.graph = g
            e.static = static
            e.outputs = out_bufs
            if self.verbose:
                print(f"[graphs] captured step for shape {key}")
            # the capture itself already executed once? No: capture does
            # not run the work; replay now for this batch.
            self._copy_into(e.static, batch)
            e.graph.replay()
            return e.outputs
        except Exception as exc:  # pragma: no cover - device dependent
            e.disabled = True
            e.graph = None
            if self.verbose:
                print(f"[graphs] capture failed for {key}: {exc}; eager")
            torch.cuda.synchronize()
            return self._fallback(batch)
