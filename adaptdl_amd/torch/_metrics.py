"""In-band step profiling feeding the goodput model.

Wall-clock step times keyed by (num_nodes, num_replicas, atomic_bsz),
separated into accumulation-step time and optimizer-step time (with the
gradient-sync share measured by the sync engine's hipEvents), fitted every
30 s on rank 0 into PerfParams and reported as sched hints.  The on-wire
hint schema and the profile-key layout are the contract shared with the
reference (``/root/reference/adaptdl/adaptdl/torch/_metrics.py``); the
implementation here is adaptdl_amd's own (sync timing comes from the
GradSyncEngine's hipEvents, not from hooks around torch DDP, and the
whole module is a singleton profiler object behind thin functions).
"""

import collections
import os
import pickle
import time

import numpy as np

import adaptdl_amd.checkpoint
import adaptdl_amd.env
from adaptdl_amd.goodput import GoodputFunction, fit_perf_params
from adaptdl_amd.sched_hints import SCHED_HINTS, PERF_PARAMS, post_sched_hints

_FIT_INTERVAL = float(os.getenv("ADAPTDL_FIT_INTERVAL", "30"))

# Counter field names inside each profile bucket (the contract the
# restart tests and the goodput fit both rely on).
_F_ACC_T, _F_ACC_N = "accum_step_time", "accum_count"
_F_OPT_T, _F_OPT_SYNC, _F_OPT_N = ("optim_step_time", "optim_sync_time",
                                   "optim_count")


class _MetricsState(adaptdl_amd.checkpoint.State):
    """Checkpointable profiler state (named slot ``adaptdl-metrics``)."""

    _FIELDS = ("profile", "perf_params", "grad_params", "init_batch_size",
               "max_batch_size", "local_bsz_bounds", "gradient_accumulation",
               "progress")

    def __init__(self):
        super().__init__("adaptdl-metrics")
        self.profile = collections.defaultdict(collections.Counter)
        self.perf_params = None
        self.grad_params = None
        self.init_batch_size = None
        self.max_batch_size = None
        self.local_bsz_bounds = None
        self.gradient_accumulation = False
        self.progress = 0.0  # scale-invariant progress (statistical steps)

    def save(self, fileobj):
        pickle.dump({f: getattr(self, f) for f in self._FIELDS}, fileobj)

    def load(self, fileobj):
        for f, v in pickle.load(fileobj).items():
            setattr(self, f, v)


class _Profiler:
    """Owns the metrics state plus the in-flight step measurement."""

    def __init__(self):
        self._state = None
        self._last_report = None
        self._adp_grad_stats = {}   # per-ADP-instance (norm_sqr, variance)
        # In-flight step measurement (None outside profile_step_start/commit).
        self._open_step = None

    @property
    def state(self):
        if self._state is None:
            self._state = _MetricsState()
            adaptdl_amd.checkpoint.load_state(self._state)
        return self._state

    # ---- step timing ----------------------------------------------------

    def begin_step(self, atomic_bsz):
        self._open_step = {"bsz": atomic_bsz, "t0": time.time(), "sync": 0.0}

    def add_sync_time(self, seconds):
        if self._open_step is not None:
            self._open_step["sync"] += seconds

    def commit_step(self, accumulation_step):
        cur, self._open_step = self._open_step, None
        elapsed = time.time() - cur["t0"]
        bucket = self.state.profile[
            (adaptdl_amd.env.num_nodes(), adaptdl_amd.env.num_replicas(),
             cur["bsz"])]
        if accumulation_step:
            bucket[_F_ACC_T] += elapsed
            bucket[_F_ACC_N] += 1
            return
        bucket[_F_OPT_T] += elapsed
        bucket[_F_OPT_SYNC] += min(cur["sync"], elapsed)
        bucket[_F_OPT_N] += 1
        self._maybe_report()

    def _maybe_report(self):
        now = time.time()
        if self._last_report is None:
            self._last_report = now
            return
        if adaptdl_amd.env.replica_rank() != 0:
            return
        if now - self._last_report > _FIT_INTERVAL:
            self.refit()
            self.push_hints()
            self._last_report = now

    # ---- gradient statistics -------------------------------------------

    def record_grad_stats(self, key, grad_norm_sqr, grad_variance):
        self._adp_grad_stats[key] = (grad_norm_sqr, grad_variance)
        norm = sum(v[0] for v in self._adp_grad_stats.values())
        var = sum(v[1] for v in self._adp_grad_stats.values())
        self.state.grad_params = (norm, var)

    # ---- fitting + reporting -------------------------------------------

    def refit(self):
        """Fit PerfParams to every profile bucket that saw an optim step."""
        state = self.state
        rows = [(k, v) for k, v in state.profile.items()
                if v.get(_F_OPT_N)]
        if not rows:
            return
        keys = np.array([k for k, _ in rows])        # (n, 3)
        cols = {f: np.array([v.get(f, 0) for _, v in rows], dtype=float)
                for f in (_F_ACC_T, _F_ACC_N, _F_OPT_T, _F_OPT_SYNC,
                          _F_OPT_N)}
        assert np.all(cols[_F_OPT_N] > 0)
        assert np.all(cols[_F_OPT_T] >= cols[_F_OPT_SYNC])
        # An optim step minus its sync share is one more accumulation-shaped
        # data point; pooling them conditions the compute-time fit better.
        pooled_t = cols[_F_ACC_T] + cols[_F_OPT_T] - cols[_F_OPT_SYNC]
        pooled_n = cols[_F_ACC_N] + cols[_F_OPT_N]
        state.perf_params = fit_perf_params(
            keys[:, 0], keys[:, 1], keys[:, 2],
            pooled_t / pooled_n, cols[_F_OPT_T] / cols[_F_OPT_N])

    def push_hints(self):
        state = self.state
        if state.perf_params is None:
            return
        hints = dict(SCHED_HINTS,
                     perfParams=dict(zip(PERF_PARAMS.keys(),
                                         state.perf_params)),
                     maxBatchSize=state.max_batch_size,
                     localBszBounds=state.local_bsz_bounds,
                     initBatchSize=state.init_batch_size,
                     maxProfiledReplicas=max(k[1] for k in state.profile),
                     gradientAccumulation=state.gradient_accumulation)
        if state.grad_params:
            hints["gradParams"] = {"norm": float(state.grad_params[0]),
                                   "var": float(state.grad_params[1])}
        post_sched_hints(hints, adaptdl_amd.env.job_id())


_PROFILER = _Profiler()


# ---- module-level API (names shared with the reference wire contract) ----

def profile_step_start(atomic_bsz):
    _PROFILER.begin_step(atomic_bsz)


def profile_sync_time(sync_time):
    _PROFILER.add_sync_time(sync_time)


def profile_step_commit(accumulation_step=False):
    _PROFILER.commit_step(accumulation_step)


def update_grad_params(key, grad_norm_sqr, grad_variance):
    """Record GNS stats per AdaptiveDataParallel instance; sums over all."""
    _PROFILER.record_grad_stats(key, grad_norm_sqr, grad_variance)


def update_progress(progress):
    _PROFILER.state.progress = progress


def get_progress():
    return _PROFILER.state.progress


def set_batch_size(init_batch_size, max_batch_size, local_bsz_bounds,
                   gradient_accumulation):
    state = _PROFILER.state
    state.init_batch_size = init_batch_size
    state.max_batch_size = max_batch_size
    state.local_bsz_bounds = local_bsz_bounds
    state.gradient_accumulation = gradient_accumulation


def get_goodput_fn():
    state = _PROFILER.state
    if state.grad_params is None or state.perf_params is None:
        return None
    return GoodputFunction(state.perf_params, state.grad_params,
                           state.init_batch_size)


def get_perf_params():
    return _PROFILER.state.perf_params


def _fit_perf_params():
    _PROFILER.refit()


def _report_sched_hints():
    _PROFILER.push_hints()


def _metrics_state():
    return _PROFILER.state


def _reset_for_tests():
    """Clear module state (used by unit tests only)."""
    global _PROFILER
    _PROFILER = _Profiler()
