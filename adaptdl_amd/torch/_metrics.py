"""In-band step profiling feeding the goodput model.

Wall-clock step times keyed by (num_nodes, num_replicas, atomic_bsz),
separated into accumulation-step time and optimizer-step time (with the
gradient-sync share measured by the sync engine's hipEvents), fitted every
30 s on rank 0 into PerfParams and reported as sched hints.  Schema and
semantics follow the reference (``/root/reference/adaptdl/adaptdl/torch/
_metrics.py``), with sync timing supplied by adaptdl_amd's own gradient
sync engine rather than by hooks around torch DDP.
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


def profile_step_start(atomic_bsz):
    state = _metrics_state()
    state.atomic_bsz = atomic_bsz
    state.step_start = time.time()
    state.sync_time = 0.0


def profile_sync_time(sync_time):
    state = _metrics_state()
    if hasattr(state, "sync_time"):  # only within an active profiled step
        state.sync_time += sync_time


_PREV_REPORT = None


def profile_step_commit(accumulation_step=False):
    global _PREV_REPORT
    state = _metrics_state()
    step_time = time.time() - state.step_start
    num_nodes = adaptdl_amd.env.num_nodes()
    num_replicas = adaptdl_amd.env.num_replicas()
    key = (num_nodes, num_replicas, state.atomic_bsz)
    if accumulation_step:
        state.profile[key]["accum_step_time"] += step_time
        state.profile[key]["accum_count"] += 1
    else:
        state.profile[key]["optim_step_time"] += step_time
        state.profile[key]["optim_sync_time"] += min(state.sync_time,
                                                     step_time)
        state.profile[key]["optim_count"] += 1
    del state.atomic_bsz
    del state.step_start
    del state.sync_time
    if not accumulation_step:
        if _PREV_REPORT is None:
            _PREV_REPORT = time.time()
        if adaptdl_amd.env.replica_rank() == 0 and \
                time.time() - _PREV_REPORT > _FIT_INTERVAL:
            _fit_perf_params()
            _report_sched_hints()
            _PREV_REPORT = time.time()


_GRAD_PARAM_DICT = {}


def update_grad_params(key, grad_norm_sqr, grad_variance):
    """Record GNS stats per AdaptiveDataParallel instance; sums over all."""
    _GRAD_PARAM_DICT[key] = np.asarray([grad_norm_sqr, grad_variance])
    total = sum(_GRAD_PARAM_DICT.values())
    _metrics_state().grad_params = (total[0], total[1])


def update_progress(progress):
    _metrics_state().progress = progress


def get_progress():
    return _metrics_state().progress


def set_batch_size(init_batch_size, max_batch_size, local_bsz_bounds,
                   gradient_accumulation):
    state = _metrics_state()
    state.init_batch_size = init_batch_size
    state.max_batch_size = max_batch_size
    state.local_bsz_bounds = local_bsz_bounds
    state.gradient_accumulation = gradient_accumulation


def get_goodput_fn():
    state = _metrics_state()
    if state.grad_params is None or state.perf_params is None:
        return None
    return GoodputFunction(state.perf_params, state.grad_params,
                           state.init_batch_size)


def get_perf_params():
    return _metrics_state().perf_params


def _fit_perf_params():
    state = _metrics_state()
    profile = {k: v for k, v in state.profile.items() if v.get("optim_count")}
    if not profile:
        return
    num_nodes, num_replicas, atomic_bsz = (
        np.array(k) for k in zip(*profile.keys()))
    accum_step_time = np.array([v.get("accum_step_time", 0.0)
                                for v in profile.values()])
    accum_count = np.array([v.get("accum_count", 0) for v in profile.values()])
    optim_step_time = np.array([v.get("optim_step_time", 0.0)
                                for v in profile.values()])
    optim_sync_time = np.array([v.get("optim_sync_time", 0.0)
                                for v in profile.values()])
    optim_count = np.array([v.get("optim_count", 0) for v in profile.values()])
    assert np.all(optim_count > 0)
    # Non-sync time of optimizer steps ~ accumulation-step time; pool them
    # for a better-conditioned compute-time fit.
    assert np.all(optim_step_time >= optim_sync_time)
    accum_step_time = accum_step_time + optim_step_time - optim_sync_time
    accum_count = accum_count + optim_count
    accum_step_time = accum_step_time / accum_count
    optim_step_time = optim_step_time / optim_count
    state.perf_params = fit_perf_params(num_nodes, num_replicas, atomic_bsz,
                                        accum_step_time, optim_step_time)


def _report_sched_hints():
    state = _metrics_state()
    if state.perf_params is None:
        return
    sched_hints = SCHED_HINTS.copy()
    sched_hints["perfParams"] = dict(zip(PERF_PARAMS.keys(),
                                         state.perf_params))
    sched_hints["maxBatchSize"] = state.max_batch_size
    sched_hints["localBszBounds"] = state.local_bsz_bounds
    sched_hints["initBatchSize"] = state.init_batch_size
    if state.grad_params:
        sched_hints["gradParams"] = {"norm": float(state.grad_params[0]),
                                     "var": float(state.grad_params[1])}
    sched_hints["maxProfiledReplicas"] = max(k[1] for k in state.profile)
    sched_hints["gradientAccumulation"] = state.gradient_accumulation
    post_sched_hints(sched_hints, adaptdl_amd.env.job_id())


class _MetricsState(adaptdl_amd.checkpoint.State):
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
        pickle.dump((self.profile, self.perf_params, self.grad_params,
                     self.init_batch_size, self.max_batch_size,
                     self.local_bsz_bounds, self.gradient_accumulation,
                     self.progress), fileobj)

    def load(self, fileobj):
        (self.profile, self.perf_params, self.grad_params,
         self.init_batch_size, self.max_batch_size, self.local_bsz_bounds,
         self.gradient_accumulation, self.progress) = pickle.load(fileobj)


_METRICS_STATE = None


def _metrics_state():
    global _METRICS_STATE
    if _METRICS_STATE is None:
        _METRICS_STATE = _MetricsState()
        adaptdl_amd.checkpoint.load_state(_METRICS_STATE)
    return _METRICS_STATE


def _reset_for_tests():
    """Clear module state (used by unit tests only)."""
    global _METRICS_STATE, _PREV_REPORT
    _METRICS_STATE = None
    _PREV_REPORT = None
    _GRAD_PARAM_DICT.clear()
