import numpy as np
import torch

import adaptdl_amd.collective as collective
import adaptdl_amd.checkpoint as checkpoint
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


def test_elastic_sampler_partition():
    import adaptdl_amd.torch.data as data_mod

    class DS(torch.utils.data.Dataset):
        def __len__(self):
            return 10

        def __getitem__(self, i):
            return i

    sampler = data_mod.ElasticSampler(DS(), shuffle=False)
    sampler.num_replicas = 3
    sampler.rank = 1
    sampler.set_epoch(0, index=0)
    idx = list(iter(sampler))
    assert idx == [1, 4, 7, 1]  # padded to ceil(10/3)=4
    sampler.rank = 0
    assert list(iter(sampler)) == [0, 3, 6, 9]
    # Resume from global index 4.
    sampler.set_epoch(0, index=4)
    assert list(iter(sampler)) == [4, 7]


def test_elastic_sampler_shuffle_deterministic():
    import adaptdl_amd.torch.data as data_mod

    class DS(torch.utils.data.Dataset):
        def __len__(self):
            return 100

        def __getitem__(self, i):
            return i

    s1 = data_mod.ElasticSampler(DS(), shuffle=True)
    s2 = data_mod.ElasticSampler(DS(), shuffle=True)
    for s in (s1, s2):
        s.num_replicas, s.rank = 2, 0
        s.set_epoch(3, index=0)
    assert list(iter(s1)) == list(iter(s2))
    s2.rank = 1
    part0, part1 = set(iter(s1)), set(iter(s2))
    assert len(part0 | part1) == 100


@elastic_multiprocessing
def _run_dataloader_restarts():
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.epoch import remaining_epochs_until
    collective.initialize()
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())

    dataset = torch.utils.data.TensorDataset(
        torch.arange(64, dtype=torch.float32).unsqueeze(1))
    loader = adl.AdaptiveDataLoader(dataset, batch_size=8, shuffle=False)
    seen = 0
    for epoch in remaining_epochs_until(2):
        for (batch,) in loader:
            seen += batch.numel() * env.num_replicas()
            if env.num_restarts() == 0 and epoch == 0 and \
                    loader._elastic.current_index >= 24:
                checkpoint.save_all_states()
                collective.teardown()
                torch.distributed.destroy_process_group()
                return 4
            if env.num_restarts() == 1 and epoch == 1 and \
                    loader._elastic.current_index >= 32:
                checkpoint.save_all_states()
                collective.teardown()
                torch.distributed.destroy_process_group()
                return 2
    assert seen > 0
    collective.teardown()
    torch.distributed.destroy_process_group()
    return 0


def _free_port():
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def test_dataloader_restarts():
    _run_dataloader_restarts()


@elastic_multiprocessing
def _run_dataloader_break():
    import adaptdl_amd.torch as adl
    from adaptdl_amd.torch.epoch import remaining_epochs_until
    collective.initialize()
    torch.distributed.init_process_group(
        "gloo", init_method="tcp://127.0.0.1:{}".format(
            collective.broadcast(_free_port())),
        world_size=env.num_replicas(), rank=env.replica_rank())
    dataset = torch.utils.data.TensorDataset(torch.arange(16.).unsqueeze(1))
    loader = adl.AdaptiveDataLoader(dataset, batch_size=4, shuffle=False)
    for epoch in remaining_epochs_until(1):
        for i, (batch,) in enumerate(loader):
            if i == 1:
                break  # break mid-loop; loop position must still advance
        for i, (batch,) in enumerate(loader):
            pass
    collective.teardown()
    torch.distributed.destroy_process_group()
    return 2 if env.num_restarts() == 0 else 0


def test_dataloader_break():
    _run_dataloader_break()
