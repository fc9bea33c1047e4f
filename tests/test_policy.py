"""Pollux policy + SpeedupFunction tests.

Mirrors the reference's policy test strategy
(/root/reference/sched/adaptdl_sched/policy/{speedup,pollux,
non_preemptible}_test.py): mocked-goodput speedup/memoization checks, and
mini-simulations over parametrized cluster shapes with realistic fitted
PerfParams asserting allocation validity.
"""

import time
from collections import Counter
from unittest.mock import Mock

import numpy as np
import pytest

from adaptdl_amd.goodput import GoodputFunction, PerfParams, GradParams
from adaptdl_amd.sched.policy import (PolluxPolicy, SpeedupFunction,
                                      JobInfo, NodeInfo)

PERF_PARAMS = PerfParams(0.121, 0.00568, 0.0236, 0.00634,
                         0.0118, 0.00317, 1.14)
GRAD_PARAMS = GradParams(sqr=0.00136, var=0.000502)


def _speedup_fn():
    goodput_fn = GoodputFunction(PERF_PARAMS, GRAD_PARAMS, 128)
    return SpeedupFunction(goodput_fn, max_batch_size=1280,
                           atomic_bsz_range=(64, 256))


def _mock_optimize(num_nodes, num_replicas, *args, **kwargs):
    return 32 * np.sqrt(num_replicas), 32, 0


def test_speedup_values():
    goodput_fn = Mock()
    goodput_fn.optimize = Mock(side_effect=_mock_optimize)
    fn = SpeedupFunction(goodput_fn)
    replicas = np.arange(1, 100)
    assert np.allclose(fn(1, replicas), np.sqrt(replicas))
    assert fn(1, 4) == pytest.approx(2.0)  # scalar in, scalar out
    assert fn(0, 0) == 0.0


def test_speedup_memoization():
    goodput_fn = Mock()
    goodput_fn.optimize = Mock(side_effect=_mock_optimize)
    fn = SpeedupFunction(goodput_fn)
    goodput_fn.optimize.reset_mock()
    n = np.arange(1, 10)
    assert np.allclose(fn(n, n), np.sqrt(n))
    assert goodput_fn.optimize.call_count == 1
    n2 = np.arange(1, 20)
    assert np.allclose(fn(n2, n2), np.sqrt(n2))
    assert goodput_fn.optimize.call_count == 2
    # only the unseen suffix should have been recomputed
    assert np.all(goodput_fn.optimize.call_args[0][0] == np.arange(10, 20))
    n3 = np.arange(5, 15)
    assert np.allclose(fn(n3, n3), np.sqrt(n3))
    assert goodput_fn.optimize.call_count == 2  # all memoized


@pytest.mark.parametrize("num_nodes", [1, 2, 4, 8, 16])
def test_optimize_valid_allocations(num_nodes, total_devices=16):
    num_devices = total_devices // num_nodes
    speedup_fn = _speedup_fn()
    now = time.time()
    job_resources = {"amd.com/gpu": 1, "pods": 1}
    jobs = {i: JobInfo(job_resources, speedup_fn, now + 60 * i,
                       min_replicas=0, max_replicas=8)
            for i in range(16)}
    node_resources = {"amd.com/gpu": num_devices, "pods": 32}
    nodes = {i: NodeInfo(node_resources, preemptible=False)
             for i in range(num_nodes)}
    template = NodeInfo(node_resources, preemptible=True)
    policy = PolluxPolicy(seed=0)
    prev_allocs = {}
    for cycle in range(2):
        allocations, desired_nodes = policy.optimize(
            jobs, nodes, prev_allocs, template)
        assert desired_nodes >= 1
        node_count = Counter()
        for job_key, placement in allocations.items():
            assert len(placement) <= jobs[job_key].max_replicas
            for node_key in placement:
                assert node_key in nodes
                node_count[node_key] += 1
        for node_key, count in node_count.items():
            assert count <= nodes[node_key].resources["amd.com/gpu"]
        # The cluster should not sit idle given 16 elastic jobs.
        assert sum(len(a) for a in allocations.values()) > 0
        prev_allocs = allocations


def test_allocate_job_first_fit():
    nodes = {
        "0": NodeInfo({"gpu": 1, "cpu": 500, "pods": 32}, preemptible=False),
        "1": NodeInfo({"gpu": 2, "cpu": 2000, "pods": 32}, preemptible=False),
        "2": NodeInfo({"gpu": 2, "cpu": 3000, "pods": 32}, preemptible=True),
    }
    speedup_fn = _speedup_fn()
    now = time.time()
    policy = PolluxPolicy()
    job_1 = JobInfo({"gpu": 1, "cpu": 500, "pods": 1}, speedup_fn, now,
                    0, max_replicas=1)
    job_2 = JobInfo({"gpu": 1, "cpu": 1000, "pods": 1}, speedup_fn, now,
                    0, max_replicas=1)
    job_3 = JobInfo({"gpu": 1, "cpu": 1000, "pods": 1}, speedup_fn, now,
                    2, max_replicas=2)
    job_4 = JobInfo({"gpu": 1, "cpu": 2000, "pods": 1}, speedup_fn, now,
                    2, max_replicas=2)
    assert policy.allocate_job(job_1, nodes) == ["0"]
    assert policy.allocate_job(job_2, nodes) == ["1"]
    assert policy.allocate_job(job_3, nodes) == ["1", "1"]
    assert policy.allocate_job(job_4, nodes) == []


def test_unusable_node_requests_scaleup():
    nodes = {i: NodeInfo({"gpu": 1, "cpu": 500 if i == 0 else 8000,
                          "pods": 32}, preemptible=False)
             for i in range(3)}
    template = NodeInfo({"gpu": 1, "cpu": 8000, "pods": 32},
                        preemptible=True)
    speedup_fn = _speedup_fn()
    now = time.time()
    jobs = {i: JobInfo({"gpu": 1, "cpu": 1000, "pods": 1}, speedup_fn,
                       now + 60 * i, 0, max_replicas=1)
            for i in range(3)}
    policy = PolluxPolicy(seed=0, pop_size=40, generations=40)
    allocations, desired_nodes = policy.optimize(jobs, nodes, {}, template)
    assert desired_nodes > 3  # node 0 is cpu-starved -> ask for more nodes
    assert max(len(a) for a in allocations.values()) == 1
    assert sum(len(a) for a in allocations.values()) == 2


def test_non_preemptible_jobs_pinned():
    speedup_fn = _speedup_fn()
    now = time.time()
    job_resources = {"amd.com/gpu": 1, "pods": 1}
    node_resources = {"amd.com/gpu": 4, "pods": 32}
    nodes = {i: NodeInfo(node_resources, preemptible=False)
             for i in range(4)}
    template = NodeInfo(node_resources, preemptible=True)
    jobs = {}
    for i in range(4):
        jobs[i] = JobInfo(job_resources, speedup_fn, now + 60 * i,
                          min_replicas=0, max_replicas=8)
    for i in range(4, 8):
        jobs[i] = JobInfo(job_resources, speedup_fn, now + 60 * i,
                          min_replicas=2, max_replicas=4, preemptible=False)
    policy = PolluxPolicy(seed=0, pop_size=40, generations=40)
    prev_allocs = {}
    for cycle in range(3):
        allocations, _ = policy.optimize(jobs, nodes, prev_allocs, template)
        node_count = Counter()
        for job_key, placement in allocations.items():
            assert len(placement) <= jobs[job_key].max_replicas
            if placement:
                assert len(placement) >= jobs[job_key].min_replicas
            for node_key in placement:
                node_count[node_key] += 1
        for node_key, count in node_count.items():
            assert count <= node_resources["amd.com/gpu"]
        # Once a non-preemptible job has an allocation, later cycles must
        # not move it.
        for i in range(4, 8):
            if prev_allocs.get(i):
                assert allocations[i] == prev_allocs[i], \
                    "pinned job {} moved".format(i)
        prev_allocs = allocations
