"""AdaptiveBPTTIterator tests.

Mirrors the reference's BPTT restart test
(/root/reference/adaptdl/adaptdl/torch/data_test.py:143-168): batch
shapes before/after a 1 -> 2 replica restart, symmetric step counts, and
full-coverage iteration on a single replica.
"""

import torch

import adaptdl_amd.checkpoint
import adaptdl_amd.collective
import adaptdl_amd.env
from adaptdl_amd.torch.iterator import AdaptiveBPTTIterator

from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_bptt_single_replica():
    adaptdl_amd.collective.initialize("127.0.0.1")
    data = torch.arange(101)
    it = AdaptiveBPTTIterator(data, batch_size=10, bptt_len=5)
    seen = []
    for text, target in it:
        assert text.shape == target.shape
        assert text.shape[1] == 10
        assert text.shape[0] <= 5
        # target is text shifted by one step in the reshaped matrix
        assert torch.equal(text[1:], target[:-1])
        seen.append(text)
    rows = sum(t.shape[0] for t in seen)
    assert rows >= 101 // 10 - 5  # covered (nearly) all reshaped rows
    adaptdl_amd.collective.teardown()
    return 0


def test_bptt_single_replica_covers_corpus():
    _run_bptt_single_replica()


@elastic_multiprocessing
def _run_bptt_restart():
    adaptdl_amd.collective.initialize("127.0.0.1")
    data = torch.arange(500)
    bptt_iter = AdaptiveBPTTIterator(data, batch_size=10, bptt_len=5)
    idx = 0
    for idx, (text, target) in enumerate(bptt_iter):
        if adaptdl_amd.env.num_restarts() == 0 and idx == 1:
            assert text.shape == (5, 10)
            adaptdl_amd.checkpoint.save_all_states()
            adaptdl_amd.collective.teardown()
            return 2
        if adaptdl_amd.env.num_replicas() == 2:
            # Fixed global batch 10 over 2 replicas -> width 5 (reference
            # data_test.py:166-168 expects (5, 5) / (4, 5) batches).
            assert text.shape[1] == 5
            assert text.shape[0] <= 5
    if adaptdl_amd.env.num_replicas() == 2:
        # Both replicas take the same (min-capped) number of steps.
        counts = adaptdl_amd.collective.allreduce(
            [idx], lambda a, b: a + b)
        assert len(set(counts)) == 1
    adaptdl_amd.collective.teardown()
    return 0


def test_bptt_restart_shapes():
    _run_bptt_restart()


def test_bptt_gradient_accumulation_flag(tmp_ckpt_env):
    """BPTT autoscaling can enable accumulation (extension over the
    reference iterator, which never passes the flag and therefore can
    never scale a single replica's batch)."""
    import adaptdl_amd.torch as adl
    import torch as _torch
    import adaptdl_amd.collective as _collective
    if not _collective.initialized():
        _collective.initialize(master_addr="127.0.0.1")
    it = adl.AdaptiveBPTTIterator(_torch.arange(1000), batch_size=4,
                                  bptt_len=10, max_batch_size=64,
                                  local_bsz_bounds=(2, 8),
                                  gradient_accumulation=True)
    assert it._elastic._gradient_accumulation is True
    assert it._elastic.max_batch_size == 64
    _collective.teardown()
