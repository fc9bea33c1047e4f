"""Property-based PolluxPolicy allocation-validity tests (hypothesis).

Randomizes the job mix (min/max replicas, non-preemptible pins) and
cluster shape and asserts the invariants the NSGA-II repair operators
must enforce (reference pollux.py:362-428): per-node GPU capacity,
per-job max_replicas, allocations only on existing nodes, and pinned
(non-preemptible) jobs keeping their base allocation.
"""

import time
from collections import Counter

import numpy as np
from hypothesis import given, settings, strategies as st

from adaptdl_amd.sched.policy import JobInfo, NodeInfo, PolluxPolicy


def _speedup_fn():
    def fn(nodes, replicas):
        return np.sqrt(np.maximum(np.asarray(replicas, dtype=float), 0))
    return fn


@settings(max_examples=10, deadline=None)
@given(data=st.data(),
       num_jobs=st.integers(min_value=1, max_value=6),
       num_nodes=st.integers(min_value=1, max_value=4),
       gpus_per_node=st.integers(min_value=1, max_value=8))
def test_random_mix_allocations_valid(data, num_jobs, num_nodes,
                                      gpus_per_node):
    now = time.time()
    jobs = {}
    for i in range(num_jobs):
        max_r = data.draw(st.integers(min_value=1, max_value=8),
                          label="max_r")
        min_r = data.draw(st.integers(min_value=0, max_value=max_r),
                          label="min_r")
        preemptible = data.draw(st.booleans(), label="preemptible")
        jobs[i] = JobInfo({"amd.com/gpu": 1, "pods": 1}, _speedup_fn(),
                          now + i, min_replicas=min_r,
                          max_replicas=max_r, preemptible=preemptible)
    nodes = {n: NodeInfo({"amd.com/gpu": gpus_per_node, "pods": 32},
                         preemptible=False) for n in range(num_nodes)}
    template = NodeInfo({"amd.com/gpu": gpus_per_node, "pods": 32},
                        preemptible=True)

    # Base allocations: give each non-preemptible job a feasible pin.
    base = {}
    free = Counter({n: gpus_per_node for n in nodes})
    for key, job in jobs.items():
        if job.preemptible:
            continue
        want = max(job.min_replicas, 1)
        for n in nodes:
            if free[n] >= want:
                base[key] = [n] * want
                free[n] -= want
                break

    policy = PolluxPolicy(seed=1, pop_size=12, generations=6)
    allocations, desired_nodes = policy.optimize(jobs, nodes, base,
                                                 template)
    assert desired_nodes >= 1
    node_count = Counter()
    for key, placement in allocations.items():
        assert len(placement) <= jobs[key].max_replicas
        for n in placement:
            assert n in nodes
            node_count[n] += 1
    for n, count in node_count.items():
        assert count <= nodes[n].resources["amd.com/gpu"]
    # pinned jobs keep their base allocation
    for key, placement in base.items():
        if not jobs[key].preemptible:
            assert sorted(allocations.get(key, [])) == sorted(placement)
