import adaptdl_amd.collective as collective
import adaptdl_amd.checkpoint as checkpoint
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_epochs():
    from adaptdl_amd.torch.epoch import (remaining_epochs_until,
                                         current_epoch, finished_epochs)
    collective.initialize()
    seen = []
    for epoch in remaining_epochs_until(6):
        assert current_epoch() == epoch
        assert finished_epochs() == epoch
        seen.append(epoch)
        if env.num_restarts() == 0 and epoch == 2:
            # Checkpoint happens DURING epoch 2 => epoch 2 replays after
            # the restart.
            checkpoint.save_all_states()
            collective.teardown()
            return 2
        if env.num_restarts() == 1 and epoch == 4:
            checkpoint.save_all_states()
            collective.teardown()
            return 3
    assert current_epoch() is None
    assert finished_epochs() == 6
    if env.num_restarts() == 0:
        assert seen == [0, 1, 2]
    elif env.num_restarts() == 1:
        assert seen[0] == 2
    else:
        assert seen[0] == 4
        assert seen == [4, 5]
        collective.teardown()
        return 0
    collective.teardown()
    return 0


def test_epoch_replay_skip():
    _run_epochs()
