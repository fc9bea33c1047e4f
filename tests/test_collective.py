import adaptdl_amd.collective as collective
import adaptdl_amd.env as env

from conftest import elastic_multiprocessing


@elastic_multiprocessing
def _run_allreduce():
    collective.initialize()
    result = collective.allreduce(env.replica_rank() + 1)
    assert result == sum(range(1, env.num_replicas() + 1))
    result = collective.allreduce(env.replica_rank(), lambda a, b: max(a, b))
    assert result == env.num_replicas() - 1
    futures = [collective.allreduce_async(i, lambda a, b: a + b)
               for i in range(10)]
    for i, fut in enumerate(futures):
        assert fut.result() == i * env.num_replicas()
    assert collective.broadcast("hello-{}".format(env.replica_rank())) \
        == "hello-0"
    collective.teardown()
    if env.num_restarts() == 0:
        return 4
    if env.num_restarts() == 1:
        return 2
    return 0


def test_allreduce_broadcast():
    _run_allreduce()


@elastic_multiprocessing
def _run_object_reduce():
    collective.initialize()

    def dict_sum(a, b):
        for k, v in b.items():
            a[k] = a.get(k, 0) + v
        return a

    result = collective.allreduce({"x": 1, "r{}".format(env.replica_rank()):
                                   env.replica_rank()}, dict_sum)
    assert result["x"] == env.num_replicas()
    collective.teardown()
    return 3 if env.num_restarts() == 0 else 0


def test_object_reduce():
    _run_object_reduce()
