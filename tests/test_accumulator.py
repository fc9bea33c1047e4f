import adaptdl_amd.collective as collective
import adaptdl_amd.checkpoint as checkpoint
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_accumulator():
    from adaptdl_amd.torch.accumulator import Accumulator
    from adaptdl_amd.torch.epoch import remaining_epochs_until
    collective.initialize()
    accum = Accumulator()
    for epoch in remaining_epochs_until(3):
        accum["count"] += 1
        accum["rank_sum"] += env.replica_rank()
        with accum.synchronized():
            assert accum["count"] >= epoch + 1
            total_count = accum["count"]
        if epoch == 1 and env.num_restarts() == 0:
            checkpoint.save_all_states()
            collective.teardown()
            return 2
    with accum.synchronized():
        assert accum["count"] == total_count
    collective.teardown()
    return 0


def test_accumulator_restart():
    _run_accumulator()


@elastic_multiprocessing
def _run_accumulator_modes():
    from adaptdl_amd.torch.accumulator import Accumulator
    collective.initialize()
    accum = Accumulator()
    accum["a"] += 5
    accum["a"] -= 2
    accum.update({"b": 1}, c=2)
    accum.subtract({"b": 1})
    n = env.num_replicas()
    with accum.synchronized():
        assert accum["a"] == 3 * n
        assert accum["b"] == 0
        assert accum["c"] == 2 * n
        assert "a" in accum
        assert len(accum) == 3
        accum.clear()
    with accum.synchronized():
        assert len(accum) == 0
    collective.teardown()
    return 2 if env.num_restarts() == 0 else 0


def test_accumulator_modes():
    _run_accumulator_modes()
