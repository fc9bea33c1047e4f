import pickle

import adaptdl_amd.checkpoint as checkpoint
import adaptdl_amd.collective as collective
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


class _CounterState(checkpoint.State):
    def __init__(self, name):
        super().__init__(name)
        self.count = 0
        self.synced = 0

    def sync(self):
        self.synced = collective.allreduce(self.count, lambda a, b: max(a, b))

    def save(self, fileobj):
        pickle.dump(self.synced, fileobj)

    def load(self, fileobj):
        self.count = pickle.load(fileobj)


@elastic_multiprocessing
def _run_ckpt():
    collective.initialize()
    state = _CounterState("counter")
    restored = checkpoint.load_state(state)
    if env.num_restarts() == 0:
        assert not restored
        assert state.count == 0
        state.count = env.replica_rank() + 10
        checkpoint.save_all_states()
        collective.teardown()
        return 3
    if env.num_restarts() == 1:
        assert restored
        # sync() took the max count across the old 1-replica group.
        assert state.count == 10
        state.count += env.replica_rank()
        checkpoint.save_all_states()
        collective.teardown()
        return 2
    assert restored
    assert state.count == 12  # max(10+0, 10+1, 10+2)
    collective.teardown()
    return 0


def test_checkpoint_restarts():
    _run_ckpt()
