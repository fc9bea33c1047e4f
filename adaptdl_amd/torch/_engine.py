"""Gradient synchronization engine: flat buckets + RCCL + fused statistics.

This replaces both torch DistributedDataParallel's bucketed reducer and the
reference's per-parameter backward hooks (reference: ``adaptdl/adaptdl/torch/
parallel.py:64-146`` and ``gradient_noise_scale.py:169-209``) with a single
engine that owns the gradient memory:

- Parameters' ``.grad`` tensors are views into persistent flat per-
  (param-group, dtype) bucket buffers, so gradient accumulation, all-reduce,
  statistics, and the fused optimizer all operate on large contiguous
  buffers with zero copies.
- As each bucket's gradients become ready during backward (tracked with
  ``register_post_accumulate_grad_hook``), the bucket's local sum-of-squares
  statistic is computed (one bandwidth-bound fused HIP kernel on MI355X)
  and its RCCL all-reduce is launched asynchronously, overlapping
  communication over xGMI with the rest of backward.
- After backward, one fused pass per bucket scales gradients by
  1/(world * accum_count) while accumulating the fp64 total sum-of-squares
  that drives the gradient noise scale — the "N1" fused kernel obligation
  of SURVEY.md §7.

Unused-parameter handling is correct by construction: every bucket is
all-reduced on every synchronized step (never-computed gradients stay zero),
so replicas can never disagree on the reduction schedule.  For models whose
parameter usage is identical across replicas but varies step-to-step
(e.g. multiple ADP instances trained alternately), ``ADAPTDL_SKIP_UNUSED_
BUCKETS=1`` skips the all-reduce and statistics launches of buckets that
produced no gradient in the current cycle.
"""

import logging
import time

import torch
import torch.distributed

from adaptdl_amd import ops

LOG = logging.getLogger(__name__)

_DEFAULT_BUCKET_CAP_MB = 32

# Monotonic optimizer-cycle counter (bumped by every engine's
# zero_grad): layers cache per-cycle derived tensors (e.g. the bf16
# weight cast) against it, since weights only change between cycles.
_cycle_serial = 0


def cycle_serial():
    return _cycle_serial


def _segment_view(seg, p):
    """View a flat bucket segment with the same memory layout as ``p``.

    For channels_last (and channels_last_3d) parameters the view is
    strided to match, so autograd's in-place accumulation into ``.grad``
    is layout-identical to the produced gradient (no NCHW<->NHWC
    conversion kernels on the hot path).  The flat buffer then simply
    holds the segment in the parameter's storage order — element order is
    irrelevant to all-reduce and the sum-of-squares statistics.
    """
    if p.dim() == 4 and not p.is_contiguous() and \
            p.is_contiguous(memory_format=torch.channels_last):
        n_, c, h, w = p.shape
        return seg.view(n_, h, w, c).permute(0, 3, 1, 2)
    if p.dim() == 5 and not p.is_contiguous() and \
            p.is_contiguous(memory_format=torch.channels_last_3d):
        n_, c, d, h, w = p.shape
        return seg.view(n_, d, h, w, c).permute(0, 4, 1, 2, 3)
    return seg.view(p.shape)


class Bucket(object):
    __slots__ = ["group_idx", "flat", "segments", "ready", "work", "prev",
                 "param_flat", "sgd_momentum", "adam_state", "used",
                 "prev_cycle_valid"]

    def __init__(self, group_idx, flat, segments):
        self.group_idx = group_idx
        self.flat = flat          # persistent gradient buffer
        self.segments = segments  # list of (param, offset, numel)
        self.ready = 0
        self.work = None          # in-flight all-reduce handle
        self.prev = None          # snapshot buffer (accumulation / GNS diff)
        self.param_flat = None    # fused-optimizer flat parameter buffer
        self.sgd_momentum = None  # fused SGD momentum (FusedSGD)
        self.adam_state = None    # fused Adam moments (FusedAdam/W)
        self.used = False         # any gradient produced this cycle
        self.prev_cycle_valid = False  # prev holds this cycle's last mb

    def ensure_prev(self):
        if self.prev is None:
            self.prev = torch.zeros_like(self.flat)
        return self.prev

    def param_views(self, flat=None):
        flat = self.flat if flat is None else flat
        return [(p, _segment_view(flat[off:off + n], p))
                for p, off, n in self.segments]


class GradSyncEngine(object):
    """See module docstring.

    Arguments:
        param_groups: list of lists of parameters (optimizer group order).
        process_group: torch.distributed process group (or None if
            single-process).
        bucket_cap_mb: soft cap on bucket size; tuned for xGMI ring
            all-reduce message sizes.
        owner: object receiving statistic callbacks:
            owner._local_stat(bucket, microbatch) -> None (accumulate local
                sum-of-squares for the newest microbatch of ``bucket``)
            owner._on_sync_done() -> None (invoked after gradients are
                fully synchronized and scaled)
    """

    def __init__(self, param_groups, owner, process_group=None,
                 bucket_cap_mb=None):
        import os
        self._owner = owner
        self._pg = process_group
        self._world = (torch.distributed.get_world_size(process_group)
                       if torch.distributed.is_initialized() else 1)
        cap_mb = bucket_cap_mb or float(os.getenv("ADAPTDL_BUCKET_CAP_MB",
                                                  _DEFAULT_BUCKET_CAP_MB))
        cap_bytes = int(cap_mb * 1024 * 1024)
        # Opt-in: skip the all-reduce (and statistics launches) of
        # buckets in which NO gradient was produced this optimizer
        # cycle.  Only valid when parameter usage is IDENTICAL across
        # replicas (e.g. multi-ADP setups alternating between models) -
        # with asymmetric usage the per-communicator collective order
        # would diverge between ranks.  Default keeps the
        # always-all-reduce correct-by-construction behavior.
        self._skip_unused = \
            os.getenv("ADAPTDL_SKIP_UNUSED_BUCKETS") == "1"
        self.require_sync = True
        self.accum_count = 0       # completed (un-synced) microbatches
        # hipGraph-capture mode (set by torch.graph_step.GraphedStepper):
        # device work proceeds normally but host-visible side effects that
        # are illegal or meaningless during stream capture are deferred —
        # no hipEvent timing records, and _on_sync_done (which host-syncs
        # via stats.cpu()) is left for the stepper to invoke after replay.
        self.graph_mode = False
        self.pending_sync_done = False
        self._callback_queued = False
        self._hooks = []
        self.buckets = []
        self._param_to_bucket = {}

        n_groups = len(param_groups)
        device = None
        for gidx, params in enumerate(param_groups):
            by_dtype = {}
            for p in params:
                if not p.requires_grad:
                    continue
                by_dtype.setdefault(p.dtype, []).append(p)
                device = p.device if device is None else device
            for dtype, plist in by_dtype.items():
                # Reverse registration order approximates backward order,
                # letting early buckets fill (and all-reduce) first.
                cur, cur_bytes = [], 0
                esize = torch.tensor([], dtype=dtype).element_size()
                for p in reversed(plist):
                    cur.append(p)
                    cur_bytes += p.numel() * esize
                    if cur_bytes >= cap_bytes:
                        self._make_bucket(gidx, dtype, cur, device)
                        cur, cur_bytes = [], 0
                if cur:
                    self._make_bucket(gidx, dtype, cur, device)
        self.device = device if device is not None else torch.device("cpu")
        self.n_groups = n_groups
        # Row 0: local (per-microbatch) sum-of-squares per group.
        # Row 1: total (synchronized mean gradient) sum-of-squares per group.
        self.stats = torch.zeros(2, n_groups, dtype=torch.float64,
                                 device=self.device)
        self._stats_work = None
        self._sync_start_ev = None
        self._sync_end_ev = None
        self._sync_start_t = None
        self.last_sync_time = 0.0
        self._use_events = self.device.type == "cuda"
        for bucket in self.buckets:
            for param, _, _ in bucket.segments:
                self._hooks.append(param.register_post_accumulate_grad_hook(
                    self._grad_ready_hook))

    # ---- construction helpers -------------------------------------------

    def _make_bucket(self, gidx, dtype, params, device):
        total = sum(p.numel() for p in params)
        flat = torch.zeros(total, dtype=dtype, device=device)
        segments = []
        offset = 0
        for p in params:
            n = p.numel()
            segments.append((p, offset, n))
            p.grad = _segment_view(flat[offset:offset + n], p)
            offset += n
        bucket = Bucket(gidx, flat, segments)
        self.buckets.append(bucket)
        for p in params:
            self._param_to_bucket[p] = bucket

    # ---- backward-time flow ---------------------------------------------

    def _grad_ready_hook(self, param):
        if not self._callback_queued:
            torch.autograd.Variable._execution_engine.queue_callback(
                self._end_of_backward)
            self._callback_queued = True
        bucket = self._param_to_bucket[param]
        bucket.ready += 1
        bucket.used = True
        if bucket.ready == len(bucket.segments):
            self._flush_bucket(bucket)

    def _flush_bucket(self, bucket):
        if not self.require_sync or bucket.work is not None:
            return
        # Local statistics must read the bucket before all-reduce mutates
        # it; launched on the current stream they are ordered before the
        # RCCL stream's reduction and overlap with remaining backward work.
        self._owner._local_stat(bucket, self.accum_count)
        if self._world > 1:
            bucket.work = torch.distributed.all_reduce(
                bucket.flat, op=torch.distributed.ReduceOp.SUM,
                group=self._pg, async_op=True)

    def _end_of_backward(self):
        self._callback_queued = False
        if not self.require_sync:
            # Accumulation-only microbatch: record its statistics and keep
            # accumulating gradients in the buckets.
            self.accum_count += 1
            for bucket in self.buckets:
                if not (self._skip_unused and not bucket.used):
                    self._owner._accum_stat(bucket)
                bucket.ready = 0
            self._owner._on_accum_done()
            return
        # Flush buckets that did not fill (unused/partially-used params).
        # accum_count is incremented only after: _flush_bucket's local
        # statistics read it as "microbatches completed BEFORE this one".
        for bucket in self.buckets:
            if not (self._skip_unused and not bucket.used):
                self._flush_bucket(bucket)
            bucket.ready = 0
        self.accum_count += 1
        if self._world > 1:
            self._stats_work = torch.distributed.all_reduce(
                self.stats[0], op=torch.distributed.ReduceOp.SUM,
                group=self._pg, async_op=True)
        if self.graph_mode:
            # Timing events cannot be recorded inside a stream capture.
            self._sync_start_ev = None
            self._sync_start_t = None
        elif self._use_events:
            self._sync_start_ev = torch.cuda.Event(enable_timing=True)
            self._sync_start_ev.record()
        else:
            self._sync_start_t = time.time()
        self._finalize()

    def _finalize(self):
        scale = 1.0 / (self._world * self.accum_count)
        for bucket in self.buckets:
            if bucket.work is not None:
                bucket.work.wait()
                bucket.work = None
            if self._skip_unused and not bucket.used:
                continue  # all-zero bucket: contributes nothing
            # Scale to the mean gradient and accumulate the total
            # sum-of-squares statistic (owner may precondition it).
            self._owner._total_stat(bucket, scale)
        if self._stats_work is not None:
            self._stats_work.wait()
            self._stats_work = None
        if self.graph_mode:
            # Defer the host-side GNS update: _on_sync_done host-syncs via
            # stats.cpu(), which is illegal during stream capture.  The
            # GraphedStepper invokes it after each replay, when the stats
            # tensor holds this step's values.  last_sync_time keeps the
            # most recent eager measurement (warmup cycles run eagerly at
            # every signature change), so the goodput fit sees a realistic
            # sync cost for replayed steps rather than 0.
            self._sync_end_ev = None
            self.pending_sync_done = True
            return
        if self._use_events:
            self._sync_end_ev = torch.cuda.Event(enable_timing=True)
            self._sync_end_ev.record()
        self._owner._on_sync_done()

    # ---- step-boundary operations ---------------------------------------

    def pull_stats(self):
        """Host copy of (local_sqr, total_sqr) per group; host-syncs.

        Also finalizes the sync-time measurement: the host sync performed by
        the .cpu() copy guarantees both hipEvents have completed, so
        elapsed_time needs no extra synchronization.
        """
        stats = self.stats.cpu().numpy()
        if self._use_events and self._sync_end_ev is not None:
            self.last_sync_time = \
                self._sync_start_ev.elapsed_time(self._sync_end_ev) / 1e3
            self._sync_end_ev = None
        elif self._sync_start_t is not None:
            self.last_sync_time = time.time() - self._sync_start_t
            self._sync_start_t = None
        return stats

    def zero_grad(self):
        """Zero all gradient buckets (and statistics) for the next step."""
        global _cycle_serial
        _cycle_serial += 1
        # Note: bucket.prev is NOT zeroed — it must persist across steps for
        # the differenced single-sample GNS estimator; the accumulation path
        # re-initializes it on the first microbatch instead.
        for bucket in self.buckets:
            bucket.flat.zero_()
            bucket.ready = 0
            bucket.work = None
            bucket.used = False
            bucket.prev_cycle_valid = False
        self.stats.zero_()
        self.accum_count = 0

    def reset_local_stats(self):
        self.stats[0].zero_()

    def save_prev(self):
        """Snapshot current (synchronized) gradients for the differenced
        estimator used when only one gradient sample is available."""
        for bucket in self.buckets:
            bucket.ensure_prev().copy_(bucket.flat)

    @property
    def world_size(self):
        return self._world

    def detach(self):
        """Remove hooks and release bucket views (for tests/teardown)."""
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
        for bucket in self.buckets:
            for p, _, _ in bucket.segments:
                p.grad = None
