"""SpeedupFunction: goodput-derived job speedup vs a single replica.

Same contract as the reference (/root/reference/sched/adaptdl_sched/policy/
speedup.py:18-70): speedup(n, r) = optimized-goodput(n, r) / goodput(1, 1),
vectorized over arrays with unique-input deduplication and a small
memoization table for repeated queries from the genetic search.
"""

import numpy as np


class SpeedupFunction(object):

    def __init__(self, goodput_fn, max_batch_size=None, atomic_bsz_range=None,
                 accumulation=False, mem_size=32):
        self._goodput_fn = goodput_fn
        self._max_batch_size = max_batch_size
        self._atomic_bsz_range = atomic_bsz_range
        self._accumulation = accumulation
        self._base_goodput, _, _ = goodput_fn.optimize(
            num_nodes=1, num_replicas=1, max_batch_size=max_batch_size,
            atomic_bsz_range=atomic_bsz_range, accumulation=accumulation)
        self._mem_size = mem_size
        # memo[n, r] = speedup for (n nodes, r replicas); -1 = unknown.
        self._memo = np.full((mem_size, mem_size), -1.0)
        self._memo[0, 0] = 0.0

    def __call__(self, num_nodes, num_replicas):
        scalar = np.isscalar(num_nodes) and np.isscalar(num_replicas)
        shape = np.broadcast(num_nodes, num_replicas).shape
        nodes = np.broadcast_to(num_nodes, shape).reshape(-1)
        replicas = np.broadcast_to(num_replicas, shape).reshape(-1)
        assert np.all(nodes >= 0) and np.all(nodes <= replicas)
        assert np.all((nodes > 0) == (replicas > 0))

        out = np.full(nodes.shape, -1.0)
        small = replicas < self._mem_size
        out[small] = self._memo[nodes[small], replicas[small]]

        todo = out < 0
        if todo.any():
            pairs = np.stack([nodes[todo], replicas[todo]])
            (un, ur), inverse = np.unique(pairs, axis=1, return_inverse=True)
            goodput, _, _ = self._goodput_fn.optimize(
                un, ur, max_batch_size=self._max_batch_size,
                atomic_bsz_range=self._atomic_bsz_range,
                accumulation=self._accumulation)
            sp = goodput / self._base_goodput
            cacheable = ur < self._mem_size
            self._memo[un[cacheable], ur[cacheable]] = sp[cacheable]
            out[todo] = sp[inverse]

        assert np.all(out >= 0)
        out = out.reshape(shape)
        return out.item() if scalar else out
