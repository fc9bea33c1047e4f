"""Scheduling dataclasses shared by the policy and allocators.

Behavioral parity with the reference's JobInfo/NodeInfo
(/root/reference/sched/adaptdl_sched/policy/utils.py:16-47).
"""


class JobInfo(object):
    """Static description of one schedulable job.

    Arguments:
        resources (dict): per-replica resource request, e.g. {"amd.com/gpu": 1}.
        speedup_fn (callable): (num_nodes, num_replicas) -> speedup (vectorized).
        creation_timestamp: orderable creation time (FIFO tie-break).
        min_replicas (int): guaranteed lower bound (0 = fully elastic).
        max_replicas (int): upper bound, >= max(min_replicas, 1).
        preemptible (bool): False pins the job to its current allocation.
    """

    def __init__(self, resources, speedup_fn, creation_timestamp,
                 min_replicas, max_replicas, preemptible=True):
        assert max_replicas > 0
        assert max_replicas >= min_replicas
        self.resources = resources
        self.speedup_fn = speedup_fn
        self.creation_timestamp = creation_timestamp
        self.min_replicas = min_replicas
        self.max_replicas = max_replicas
        self.preemptible = preemptible


class NodeInfo(object):
    """Available resources on one node (for a single 8xMI355X box this is
    {"amd.com/gpu": 8}); ``preemptible`` marks spot/scale-down candidates."""

    def __init__(self, resources, preemptible):
        self.resources = resources
        self.preemptible = preemptible
