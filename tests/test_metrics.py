"""Metrics/profiling state tests.

Mirrors the reference's _metrics_test.py:21-60: per-(nodes, replicas,
atomic_bsz) profile entries, accum vs optim accounting, and survival of
the profile across checkpoint-restart.
"""

import pytest

import adaptdl_amd.checkpoint
from adaptdl_amd.env import num_restarts

from conftest import elastic_multiprocessing


@pytest.mark.parametrize("num_replicas", [1, 2])
def test_profile_restart(num_replicas):
    @elastic_multiprocessing
    def run():
        from adaptdl_amd.torch._metrics import (
            profile_step_start, profile_sync_time, profile_step_commit,
            _metrics_state)
        if num_restarts() == 0:
            profile = _metrics_state().profile
            assert len(profile) == 0
            # start a step at local_bsz=1 but never commit it
            profile_step_start(1)
            profile_sync_time(1.0)
            # a committed optim step at local_bsz=2 (sync time is
            # clamped to the wall step time, so sleep past it)
            profile_step_start(2)
            import time
            time.sleep(0.02)
            profile_sync_time(0.005)
            profile_sync_time(0.005)
            profile_step_commit()
            # an accumulation step at the same key
            profile_step_start(2)
            profile_step_commit(accumulation_step=True)
            profile = _metrics_state().profile
            key = (1, 1, 2)
            assert len(profile) == 1
            assert profile[key]["optim_count"] == 1
            assert profile[key]["accum_count"] == 1
            assert profile[key]["optim_sync_time"] == pytest.approx(0.01)
            assert profile[key]["optim_step_time"] > 0.01
            adaptdl_amd.checkpoint.save_all_states()
            return num_replicas
        else:
            profile = _metrics_state().profile
            key = (1, 1, 2)
            assert len(profile) == 1
            assert profile[key]["optim_count"] == 1
            assert profile[key]["accum_count"] == 1
            assert profile[key]["optim_sync_time"] == pytest.approx(0.01)
            # new entries accumulate under the new replica count
            profile_step_start(3)
            profile_step_commit()
            key2 = (1, num_replicas, 3)
            assert profile[key2]["optim_count"] == 1
            return 0

    run()
