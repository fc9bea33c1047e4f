"""Trainer -> allocator hint schema and transport.

Same wire schema as the reference (``/root/reference/adaptdl/adaptdl/
sched_hints.py:30-59``) so dashboards/tools reading hints keep working, but
the default transport is in-process: hints are handed to the local
``adaptdl_amd.sched`` allocator via a registry.  If ``ADAPTDL_SUPERVISOR_URL``
is set, hints are POSTed over HTTP instead (multi-job supervisor mode).
"""

import logging

LOG = logging.getLogger(__name__)

SCHED_HINTS = {
    "perfParams": None,
    "maxBatchSize": None,
    "localBszBounds": None,
    "initBatchSize": None,
    "gradParams": None,
    "maxProfiledReplicas": None,
    "gradientAccumulation": False,
}

PERF_PARAMS = {
    "alpha_c": None,
    "beta_c": None,
    "alpha_n": None,
    "beta_n": None,
    "alpha_r": None,
    "beta_r": None,
    "gamma": None,
}

# In-process hint sink: job_key -> latest hints dict.  The in-process
# allocator (adaptdl_amd.sched.allocator) reads this directly.
_LOCAL_HINTS = {}
_CALLBACKS = []


def register_hints_callback(fn):
    """Register fn(job_key, hints) invoked on every hint report."""
    _CALLBACKS.append(fn)


def get_local_hints(job_key=None):
    if job_key is None:
        return dict(_LOCAL_HINTS)
    return _LOCAL_HINTS.get(job_key)


def post_sched_hints(sched_hints, job_key):
    """Report hints to the allocator (in-process, or HTTP if configured)."""
    from adaptdl_amd import env
    _LOCAL_HINTS[job_key] = sched_hints
    for fn in list(_CALLBACKS):
        try:
            fn(job_key, sched_hints)
        except Exception:  # noqa: BLE001
            LOG.exception("sched hints callback failed")
    url = env.supervisor_url()
    if url:
        try:
            import json
            import urllib.request
            req = urllib.request.Request(
                "{}/hints/{}".format(url, job_key),
                data=json.dumps(sched_hints, default=float).encode(),
                headers={"Content-Type": "application/json"}, method="PUT")
            urllib.request.urlopen(req, timeout=5)
        except Exception:  # noqa: BLE001
            LOG.warning("failed to report hints to supervisor %s", url)
