"""hipGraph capture of the steady-state training microbatch cycle.

Hardware-validated in round 2: the flagship bench's steady-state GPU
idle drops 19% -> 1.7% under replay (profiles/r2_gaps_*), worth
~0.6-0.9 ms of the 43 ms step; engaged by default for single-GPU bench
runs of the conv workloads (`maybe_graphed_stepper(default_on=...)`,
ADAPTDL_HIPGRAPH=0/1 overrides).  Attention stacks measured worse or
unstable under capture (SDPA loses its flash backend; see
profiles/README.md r2 passes l/m) and keep the eager path.

Scope and shape of the capture
------------------------------

One optimizer-step cycle with ``A`` gradient-accumulation microbatches has
``A + 1`` microbatch positions whose *device* work differs (the engine's
fused statistic kernels depend on the position — see
``gradient_noise_scale.py`` ``_accum_stat``/``_local_stat``):

==========  ====================================================
kind        captured device work
==========  ====================================================
"first"     bucket+stats zeroing, forward+backward,
            per-bucket ``sqsum`` + prev snapshot       (pos 0)
"mid"       forward+backward, per-bucket ``sqsum_diff``
                                                (pos 1 .. A-1)
"sync"      forward+backward, local-stat flush, bucket
            all-reduce (world>1), scale + total-``sqsum``
                                                       (pos A)
"solo"      the whole cycle when A == 0 (zeroing + sync)
==========  ====================================================

Each kind is captured once (lazily, at its first occurrence after one
eager warmup cycle) into its own hipGraph; all graphs share one memory
pool and are replayed strictly in cycle order.  Everything *host-side*
stays eager: the optimizer step (patched LR scaling + fused kernels),
the GNS host math (``_on_sync_done`` — deferred by the engine's
``graph_mode`` flag and invoked here after the sync-position replay,
when the stats tensor holds the step's values), and the dataloader's
profile/exit-flag machinery, which all run between replays exactly as
they do between eager microbatches.

Because graph replay executes no Python, the host-visible engine state a
replayed microbatch would have produced (``accum_count``, zero-grad
bookkeeping, sync flags) is applied explicitly after each replay from
the known cycle position.  The values are *absolute* (not increments),
so the same code path is correct both for true hipGraph replays (no
Python ran) and for the eager test backend (Python ran and already
advanced the state to the same values).

Safety rules (all checked per cycle; violation falls back to eager):

- the cycle signature — (atomic batch size, accumulation steps, input
  shapes/dtypes) — must match the captured one; a change (e.g. the
  goodput model re-choosing the batch size) drops the graphs and
  re-warms,
- the observed sync position must match the expected one,
- AMP GradScaler (mp_scaler) is not supported (its inf-check host logic
  inspects gradients between backward and step); Adam-preconditioned
  GNS is supported with the fused optimizers (whose statistic kernels
  read the bias-correction scalars from device memory) but refused for
  stock torch.optim.Adam (per-step scalars would be baked into the
  capture),
- ``ADAPTDL_SKIP_UNUSED_BUCKETS`` is refused (usage-conditional
  collectives contradict a static capture),
- any capture error permanently disables the stepper for the run (the
  current microbatch is re-run eagerly after a clean cycle reset).

Reference note: the reference has no equivalent (it delegates launch
scheduling to torch DDP); this is an MI355X-native addition.
"""

import contextlib
import logging
import os

import torch

from adaptdl_amd.torch.data import current_dataloader

__all__ = ["GraphedStepper", "maybe_graphed_stepper"]

LOG = logging.getLogger(__name__)


class HipGraphBackend(object):
    """Real stream-capture backend (requires a GPU).

    ``capture`` records the callable's kernels into a hipGraph WITHOUT
    executing them (host-side Python runs once, recording); the caller
    replays immediately afterwards so the microbatch happens exactly
    once.  All graphs share one memory pool; replays must follow the
    capture order.
    """

    def __init__(self):
        self._pool = torch.cuda.graph_pool_handle()
        self._side = torch.cuda.Stream()

    def begin_generation(self):
        """Start a fresh capture generation (after graphs were dropped).

        Re-capturing into the memory pool of destroyed graphs trips a
        HIPCachingAllocator internal assert (use_count > 0, observed on
        ROCm 7.2 in the round-2 A/B); a new pool per generation avoids
        reusing the stale pool bookkeeping.  The old graphs must be
        garbage-collected first so their pool memory is released.
        """
        import gc
        gc.collect()
        torch.cuda.synchronize()
        self._pool = torch.cuda.graph_pool_handle()

    @contextlib.contextmanager
    def warmup(self):
        """Eager warmup on a side stream (the torch.cuda.graph protocol:
        lazy inits must not land on the capture stream)."""
        self._side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._side):
            yield
        torch.cuda.current_stream().wait_stream(self._side)

    def capture(self, fn):
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self._pool):
            out = fn()
        return graph, out

    def replay(self, graph):
        graph.replay()


class EagerBackend(object):
    """Test backend: "capture" stores the callable, "replay" executes it.

    The capture-then-immediately-replay flow therefore executes the
    microbatch exactly once, like the real backend — but through the
    full Python path with ``engine.graph_mode`` set, exercising the
    deferred-host-work and absolute-state bookkeeping on CPU.
    """

    @contextlib.contextmanager
    def warmup(self):
        yield

    def begin_generation(self):
        pass

    def capture(self, fn):
        return fn, None

    def replay(self, fn):
        fn()


class GraphedStepper(object):
    """Routes training microbatches through captured hipGraphs.

    Arguments:
        adp: the AdaptiveDataParallel instance.
        optimizer: the (patched) optimizer; its ``step()`` stays eager
            and must be called by the user after each microbatch, as in
            a plain training loop.
        fwd_bwd: callable ``fwd_bwd(*tensors) -> loss`` performing ONE
            microbatch: ``optimizer.zero_grad()``, forward, loss,
            ``loss.backward()`` — and nothing else.
        backend: capture backend; defaults to HipGraphBackend on GPU.
        warmup_cycles: eager cycles run before capturing (after enable
            and after every signature change).

    Usage::

        stepper = GraphedStepper(adp, optim, fwd_bwd)
        for batch in loader:
            stepper.microbatch(batch)   # replaces fwd_bwd(batch)
            optim.step()
    """

    def __init__(self, adp, optimizer, fwd_bwd, backend=None,
                 warmup_cycles=1):
        from adaptdl_amd.torch.gradient_noise_scale import \
            AdamGradientNoiseScale
        if getattr(adp.gns, "_mp_scaler", None) is not None:
            raise ValueError("GraphedStepper does not support mp_scaler "
                             "(GradScaler host logic is not capturable)")
        if isinstance(adp.gns, AdamGradientNoiseScale) and \
                not hasattr(optimizer, "attach_engine"):
            # With the FUSED Adam optimizers the preconditioned
            # statistics run one graph-safe kernel per bucket whose
            # bias-correction scalars live in device memory (updated by
            # the eager optimizer step between replays), so capture is
            # supported.  The stock torch.optim.Adam path bakes the
            # host-read step count into per-segment kernels and stays
            # refused.
            raise ValueError("GraphedStepper supports Adam-"
                             "preconditioned GNS only with the fused "
                             "optimizers (FusedAdam/FusedAdamW); stock "
                             "torch.optim.Adam bakes per-step bias "
                             "correction into captured kernels")
        if getattr(adp.gns.engine, "_skip_unused", False):
            # Usage-conditional collectives contradict a static capture:
            # a replayed cycle always re-issues the kernels of its
            # capture, regardless of which parameters produced
            # gradients this time.
            raise ValueError("GraphedStepper is incompatible with "
                             "ADAPTDL_SKIP_UNUSED_BUCKETS=1")
        self._adp = adp
        self._gns = adp.gns
        self._engine = adp.gns.engine
        self._optimizer = optimizer
        self._fwd_bwd = fwd_bwd
        self._backend = backend or HipGraphBackend()
        self._warmup_cycles = warmup_cycles
        self._disabled = False

        self._sig = None          # captured cycle signature
        self._accum = 0           # A of the captured cycle
        self._pos = 0             # position within the current cycle
        self._broken = False      # desynced: eager until cycle boundary
        self._warm_left = warmup_cycles
        self._graphs = {}         # kind -> (graph, loss_out)
        self._static = None       # static input buffers
        self.stats = {"captures": 0, "replays": 0, "eager": 0,
                      "fallbacks": 0}
        # Pre-allocate the differenced-estimator snapshots so they are
        # plain allocator memory, not graph-pool memory.
        for bucket in self._engine.buckets:
            bucket.ensure_prev()

    # ---- public entry ----------------------------------------------------

    def microbatch(self, *tensors):
        """Run one training microbatch (replayed when possible).

        Returns the loss tensor when available (eager runs and real
        captures return the static loss buffer; replays return the same
        buffer, which holds the value after the replay completes).
        """
        dl = current_dataloader()
        if self._disabled or dl is None or not dl.training:
            return self._run_eager(*tensors)

        accum = dl.accumulation_steps
        is_sync = dl.is_optim_step()
        shapes = tuple((tuple(t.shape), t.dtype) for t in tensors)
        if self._broken:
            # A desync ruined the current cycle: run the rest of it eager
            # and realign at the next observed cycle boundary.  The
            # captured graphs are KEPT — the signature did not change, so
            # the next full cycle replays without re-warming.
            if is_sync:
                self._broken = False
                self._pos = 0
            return self._run_eager(*tensors)
        if self._pos == 0:
            sig = (dl.current_local_bsz, accum, shapes)
            if sig != self._sig:
                self._reset(sig, accum)
        # Desync guards: position math must agree with the dataloader, and
        # every microbatch's tensors must match the captured signature (a
        # mid-cycle shape change — e.g. a short final batch with
        # drop_last=False — would otherwise hit the static-buffer copy with
        # a non-broadcastable shape).  Violations run eager until the
        # cycle boundary, keeping the graphs for the next aligned cycle.
        expected_sync = (self._accum == 0) or (self._pos == self._accum)
        if is_sync != expected_sync or accum != self._accum \
                or shapes != self._sig[2]:
            LOG.warning("graph stepper desync (pos %d, accum %d->%d, "
                        "sync %s, shape change %s); eager until the next "
                        "cycle boundary", self._pos, self._accum, accum,
                        is_sync, shapes != self._sig[2])
            self.stats["fallbacks"] += 1
            if is_sync:
                self._pos = 0     # boundary observed: already realigned
            else:
                self._broken = True
            return self._run_eager(*tensors)

        kind = ("solo" if self._accum == 0 else
                "sync" if is_sync else
                "first" if self._pos == 0 else "mid")

        if self._warm_left > 0:
            with self._backend.warmup():
                loss = self._run_eager(*tensors)
            if is_sync:
                self._warm_left -= 1
        elif kind not in self._graphs:
            loss = self._capture(kind, *tensors)
        else:
            loss = self._replay(kind, *tensors)

        self._pos = 0 if is_sync else self._pos + 1
        return loss

    # ---- internals -------------------------------------------------------

    def _run_eager(self, *tensors):
        self._engine.graph_mode = False
        self.stats["eager"] += 1
        # Replays run no Python, so cycle-keyed layer caches (e.g. the
        # bf16 weight cast) were last refreshed at capture time; bump
        # the serial so this eager microbatch re-derives them from the
        # CURRENT weights instead of a stale snapshot.
        from adaptdl_amd.torch import _engine as _engine_mod
        _engine_mod._cycle_serial += 1
        return self._fwd_bwd(*tensors)

    def _reset(self, sig, accum):
        had_graphs = bool(self._graphs)
        self._graphs.clear()
        self._static = None
        if had_graphs:
            # Fresh memory pool for the next capture generation (the old
            # graphs' pool cannot be safely re-captured into).
            self._backend.begin_generation()
        self._sig = sig
        self._accum = accum
        self._warm_left = self._warmup_cycles
        self._pos = 0
        self._broken = False

    def _ensure_static(self, tensors):
        if self._static is None:
            self._static = [torch.empty_like(t) for t in tensors]
        for dst, src in zip(self._static, tensors):
            dst.copy_(src, non_blocking=True)

    def _set_pre_state(self, kind):
        sync = kind in ("sync", "solo")
        self._adp.require_backward_grad_sync = sync
        self._engine.require_sync = sync
        self._engine.graph_mode = True

    def _apply_post_state(self, kind):
        """Absolute host state after a capture/replay of ``kind``.

        Graph replay executes no Python, so the engine's host-visible
        bookkeeping is applied here; on the eager backend the Python
        path already advanced it to these same values.
        """
        engine = self._engine
        engine._callback_queued = False
        for bucket in engine.buckets:
            bucket.ready = 0
        if kind == "first":
            engine.accum_count = 1
            self._gns._should_zero_grad = False
        elif kind == "mid":
            engine.accum_count = self._pos + 1
            self._gns._should_zero_grad = False
        else:  # sync / solo
            engine.accum_count = \
                self._accum + 1 if kind == "sync" else 1
            for bucket in engine.buckets:
                bucket.work = None
            # last_sync_time keeps the most recent eager-measured value
            # (from the warmup cycles of this signature) so the goodput
            # fit does not see all-reduce as free under replays.
            engine.pending_sync_done = False
            engine.graph_mode = False
            # Deferred host finalize: GNS estimates, scaling-rule gain,
            # progress, grad-param reporting (reads the stats tensor the
            # replay just produced).
            self._gns._on_sync_done()

    def _capture(self, kind, *tensors):
        self._ensure_static(tensors)
        self._set_pre_state(kind)
        static = self._static
        try:
            graph, out = self._backend.capture(
                lambda: self._fwd_bwd(*static))
        except Exception:
            LOG.exception("hipGraph capture failed for %r; disabling "
                          "graph stepping for this run", kind)
            self._disabled = True
            self._engine.graph_mode = False
            self.stats["fallbacks"] += 1
            # The aborted capture may have left partial side effects
            # (accum_count increments, statistic-tensor contributions).
            # Restart the cycle from clean state before re-running
            # eagerly so no microbatch's statistics are double-counted.
            # Earlier microbatches of this cycle lose their gradient
            # contribution to this one step (a once-per-run event).
            self._gns.reset_accumulation()
            self._pos = 0
            self._sig = None
            return self._run_eager(*tensors)
        self._graphs[kind] = (graph, out)
        self.stats["captures"] += 1
        # Capture records without executing: replay immediately so this
        # microbatch happens exactly once, on the data just copied in.
        self._backend.replay(graph)
        self._apply_post_state(kind)
        return out

    def _replay(self, kind, *tensors):
        self._ensure_static(tensors)
        self._set_pre_state(kind)
        graph, out = self._graphs[kind]
        self._backend.replay(graph)
        self.stats["replays"] += 1
        self._apply_post_state(kind)
        return out


def maybe_graphed_stepper(adp, optimizer, fwd_bwd, default_on=False):
    """Build a GraphedStepper iff ``ADAPTDL_HIPGRAPH=1`` (or unset with
    ``default_on``) and a GPU is available; returns None (caller keeps
    the eager path) otherwise."""
    flag = os.getenv("ADAPTDL_HIPGRAPH", "1" if default_on else "0")
    if flag != "1":
        return None
    if not torch.cuda.is_available():
        LOG.warning("ADAPTDL_HIPGRAPH=1 ignored: no GPU available")
        return None
    try:
        return GraphedStepper(adp, optimizer, fwd_bwd)
    except (ValueError, RuntimeError) as exc:
        # ValueError: unsupported configuration (mp_scaler / Adam GNS).
        # RuntimeError: HipGraphBackend construction (graph_pool_handle,
        # Stream) can fail on misconfigured ROCm setups.
        LOG.warning("ADAPTDL_HIPGRAPH=1 ignored: %s", exc)
        return None
