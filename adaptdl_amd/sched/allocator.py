"""In-process allocator: Pollux policy over the local node(s).

Single-node counterpart of the reference's allocator daemon
(/root/reference/sched/adaptdl_sched/allocator.py:56-278): builds
JobInfos from reported sched hints (GoodputFunction -> SpeedupFunction,
including the 2x maxProfiledReplicas growth rule of allocator.py:186),
runs PolluxPolicy.optimize over the node inventory, and hands the chosen
replica allocations to the controller.  There is no Kubernetes: nodes
default to the one MI355X box we run on ({"amd.com/gpu": N}).
"""

import logging
import time

from adaptdl_amd.goodput import GoodputFunction, PerfParams, GradParams
from adaptdl_amd.sched.policy import (PolluxPolicy, SpeedupFunction,
                                      JobInfo, NodeInfo)

LOG = logging.getLogger(__name__)

GPU_RESOURCE = "amd.com/gpu"


def job_info_from_hints(hints, creation_timestamp, min_replicas=0,
                        max_replicas=None, preemptible=True,
                        resources=None):
    """Build a JobInfo from a trainer's sched-hints dict (or None)."""
    resources = resources or {GPU_RESOURCE: 1}
    max_replicas = max_replicas or 64
    speedup_fn = None
    if hints and hints.get("perfParams"):
        perf_params = PerfParams(**{k: float(v) for k, v
                                    in hints["perfParams"].items()})
        if hints.get("gradParams"):
            grad_params = GradParams(
                sqr=float(hints["gradParams"]["norm"]),
                var=float(hints["gradParams"]["var"]))
        else:
            grad_params = GradParams(sqr=1.0, var=0.0)
        goodput_fn = GoodputFunction(perf_params, grad_params,
                                     int(hints["initBatchSize"]))
        speedup_fn = SpeedupFunction(
            goodput_fn,
            hints.get("maxBatchSize"),
            tuple(hints["localBszBounds"])
            if hints.get("localBszBounds") else None,
            bool(hints.get("gradientAccumulation", False)))
        # Grow replicas at most 2x past what has been profiled, so the
        # speedup model is never extrapolated too far.
        profiled = int(hints.get("maxProfiledReplicas") or 1)
        max_replicas = min(max_replicas, max(2 * profiled, 1))
    if speedup_fn is None:
        def speedup_fn(n, r):  # noqa: E306 - default: no measured speedup
            import numpy as np
            return np.minimum(np.asarray(n, dtype=float) * 0 + 1.0,
                              np.asarray(r, dtype=float) * 0 + 1.0)
        max_replicas = max(min_replicas, 1)
    return JobInfo(resources, speedup_fn, creation_timestamp,
                   min_replicas, max_replicas, preemptible)


class LocalAllocator(object):
    """Periodic Pollux optimization over local jobs.

    Arguments:
        nodes: dict node-name -> NodeInfo; defaults to one node with
            ``num_gpus`` GPUs.
        num_gpus: GPU count for the default single-node inventory.
    """

    def __init__(self, nodes=None, num_gpus=8, policy=None):
        if nodes is None:
            nodes = {"local": NodeInfo({GPU_RESOURCE: num_gpus,
                                        "pods": 64},
                                       preemptible=False)}
        self._nodes = nodes
        self._policy = policy or PolluxPolicy()
        self._template = NodeInfo(
            dict(next(iter(nodes.values())).resources), preemptible=True)

    def allocate_new(self, job_info):
        """First-fit allocation for a newly submitted job."""
        return self._policy.allocate_job(job_info, self._nodes)

    def optimize(self, jobs, base_allocations):
        """One allocation cycle.

        Arguments:
            jobs: dict job-key -> JobInfo.
            base_allocations: dict job-key -> list of node names.

        Returns:
            dict job-key -> list of node names (one per replica).
        """
        if not jobs:
            self.desired_nodes = min(1, len(self._nodes))
            return {}
        t0 = time.time()
        allocations, desired_nodes = self._policy.optimize(
            jobs, self._nodes, base_allocations, self._template)
        LOG.info("allocator cycle: %d jobs in %.2fs -> %s (desired nodes "
                 "%d)", len(jobs), time.time() - t0,
                 {k: len(v) for k, v in allocations.items()}, desired_nodes)
        # Cluster-expander signal (reference cluster_expander.py creates
        # placeholder pods so the k8s autoscaler grows the cluster; the
        # local analog is an observable target an external provisioner
        # can consume via the controller status/metrics).
        self.desired_nodes = desired_nodes
        return allocations
