"""Pollux scheduling policy: goodput-optimizing replica allocation.

Behavioral parity with the reference PolluxPolicy
(/root/reference/sched/adaptdl_sched/policy/pollux.py:34-223): a
two-objective genetic search over replica-assignment matrices
(jobs x 2*nodes, the second half being virtual nodes for cluster
autoscaling) with objectives (-total scaled speedup, cluster size),
dominant-resource-share speedup scaling, a 10% restart penalty, and a
0.35-0.65 cluster-utilization band driving the desired node count.

The implementation is self-contained: a small NSGA-II (fast
non-dominated sort + crowding distance, binary tournament, the same
crossover/mutation/repair semantics) written directly in numpy — the
reference depends on pymoo for this.  For the in-process single-node
MI355X deployment (8 GPUs, one NodeInfo) the same policy object is used
by adaptdl_amd.sched.allocator; the genetic machinery degenerates
gracefully since the state space is tiny.
"""

import copy
import logging
from collections import OrderedDict

import numpy as np

LOG = logging.getLogger(__name__)

_POP_SIZE = 100
_N_GEN = 100
_RESTART_PENALTY = 0.1


class PolluxPolicy(object):
    def __init__(self, seed=None, pop_size=_POP_SIZE, generations=_N_GEN):
        self._pop_size = pop_size
        self._generations = generations
        self._prev_population = None
        self._prev_jobs = None
        self._prev_nodes = None
        self._min_util = 0.35
        self._max_util = 0.65
        self._rng = np.random.default_rng(seed)

    # ---- single-job first-fit (new job admission) -----------------------

    def allocate_job(self, job_info, nodes):
        """First node with room for ``min_replicas`` replicas (>=1)."""
        want = max(job_info.min_replicas, 1)
        for name, node in self._sorted_nodes(nodes).items():
            fits = min((node.resources.get(rt, 0) // amount
                        for rt, amount in job_info.resources.items()
                        if amount > 0), default=0)
            if fits >= want:
                return [name] * want
        return []

    @staticmethod
    def _sorted_nodes(nodes):
        # Non-preemptible nodes first, then by name.
        return OrderedDict(sorted(nodes.items(),
                                  key=lambda kv: (kv[1].preemptible, kv[0])))

    # ---- state <-> allocation conversion --------------------------------

    @staticmethod
    def _to_state(allocations, jobs, nodes):
        jidx = {k: i for i, k in enumerate(jobs)}
        nidx = {k: i for i, k in enumerate(nodes)}
        state = np.zeros((len(jobs), len(nodes)), dtype=np.int64)
        for job_key, alloc in allocations.items():
            if job_key not in jidx:
                continue
            for node_key in alloc:
                if node_key in nidx:
                    state[jidx[job_key], nidx[node_key]] += 1
        return state

    @staticmethod
    def _to_allocations(state, jobs, nodes):
        out = {}
        node_keys = list(nodes)
        for j, job_key in enumerate(jobs):
            alloc = []
            for n, count in enumerate(state[j][:len(node_keys)]):
                alloc.extend([node_keys[n]] * int(count))
            out[job_key] = alloc
        return out

    def _warm_start(self, jobs, nodes, width):
        """Map the previous population onto the current jobs/nodes."""
        pop = np.zeros((len(self._prev_population), len(jobs), width),
                       dtype=np.int64)
        src = [i for i, k in enumerate(self._prev_jobs) if k in jobs]
        dst = [i for i, k in enumerate(jobs) if k in self._prev_jobs]
        prev_nidx = {k: i for i, k in enumerate(self._prev_nodes)}
        spare = len(self._prev_nodes)  # next previous-virtual column
        prev_width = self._prev_population.shape[2]
        for i, key in enumerate(nodes):
            if key in prev_nidx:
                pop[:, dst, i] = self._prev_population[:, src,
                                                       prev_nidx[key]]
            elif spare < prev_width:
                pop[:, dst, i] = self._prev_population[:, src, spare]
                spare += 1
        for i in range(len(nodes), width):
            if spare >= prev_width:
                break
            pop[:, dst, i] = self._prev_population[:, src, spare]
            spare += 1
        return pop

    # ---- main optimization cycle ----------------------------------------

    def optimize(self, jobs, nodes, base_allocations, node_template):
        """One allocation cycle.

        Returns (allocations, desired_nodes): a dict from job key to a
        list of node keys (one per replica), and the target cluster size
        for the autoscaler.
        """
        def pinned(key, job):
            return (not job.preemptible and
                    bool(base_allocations.get(key, [])))

        jobs = OrderedDict(sorted(
            jobs.items(), key=lambda kv: (not pinned(kv[0], kv[1]),
                                          kv[1].min_replicas,
                                          kv[1].creation_timestamp)))
        nodes = self._sorted_nodes(nodes)
        width = 2 * len(nodes)  # physical + virtual (autoscaled) nodes
        base = np.concatenate([self._to_state(base_allocations, jobs, nodes),
                               np.zeros((len(jobs), len(nodes)), np.int64)],
                              axis=1)

        if self._prev_population is None:
            seeds = base[None]
        else:
            seeds = self._warm_start(jobs, nodes, width)

        problem = _AllocationProblem(
            list(jobs.values()),
            list(nodes.values()) + [node_template] * len(nodes),
            base, self._rng)
        population, values = _nsga2(problem, seeds, self._pop_size,
                                    self._generations)

        self._prev_population = population.copy()
        self._prev_jobs = list(jobs)
        self._prev_nodes = list(nodes)

        front = _nondominated_front(values)
        population, values = population[front], values[front]
        utilities = problem.cluster_utilities(population)
        desired = self._desired_nodes(utilities, values, nodes)
        idx = self._select(values, min(len(nodes), desired))
        if LOG.isEnabledFor(logging.DEBUG):
            for i, state in enumerate(population):
                LOG.debug("solution %d: %s value=%s utility=%.3f", i,
                          state.tolist(), values[i].tolist(), utilities[i])
        if idx is None:
            return {}, desired
        return self._to_allocations(population[idx], jobs, nodes), desired

    @staticmethod
    def _select(values, max_nodes):
        """Best total-speedup solution using at most ``max_nodes`` nodes."""
        ok = values[:, 1] <= max_nodes
        if not ok.any():
            return None
        masked = np.where(ok, values[:, 0], 0.0)  # objective 0 is <= 0
        return int(np.argmin(masked))

    def _desired_nodes(self, utilities, values, nodes):
        idx = self._select(values, len(nodes))
        if idx is not None and \
                self._min_util <= utilities[idx] <= self._max_util:
            return len(nodes)
        target = (self._min_util + self._max_util) / 2
        best_util, best_nodes = np.inf, len(nodes)
        for util, (_, num_nodes) in zip(utilities, values):
            if util < self._min_util:
                continue
            if np.isclose(util, best_util) and num_nodes > best_nodes:
                best_nodes = num_nodes
            if abs(util - target) < abs(best_util - target):
                best_util, best_nodes = util, num_nodes
        return int(best_nodes)


class _AllocationProblem(object):
    """Objectives + genetic operators over (num_jobs x width) states."""

    def __init__(self, jobs, nodes, base, rng):
        assert base.shape == (len(jobs), len(nodes))
        self.jobs = jobs
        self.nodes = nodes
        self.base = base
        self.rng = rng
        self.pinned = [j for j, job in enumerate(jobs)
                       if not job.preemptible and base[j].any()]

        rtypes = sorted(set().union(*[set(j.resources) for j in jobs]))
        self.job_res = np.array(
            [[job.resources.get(rt, 0) for rt in rtypes] for job in jobs],
            dtype=np.int64)
        self.node_res = np.array(
            [[node.resources.get(rt, 0) for rt in rtypes] for node in nodes],
            dtype=np.int64)
        total = np.sum(self.node_res, axis=0)
        with np.errstate(divide="ignore", invalid="ignore"):
            shares = np.where(total > 0,
                              self.job_res / np.maximum(total, 1), 0.0)
        self.dominant_share = np.amax(shares, axis=1)

        # Per-(job, node) replica caps, after subtracting pinned usage.
        avail = self.node_res.astype(np.int64).T.copy()  # (rtypes, nodes)
        for j in self.pinned:
            avail -= np.outer(self.job_res[j], base[j])
        assert (avail >= 0).all()
        self.cap = np.zeros_like(base)
        for j, job in enumerate(jobs):
            used = self.job_res[j] > 0
            with np.errstate(divide="ignore"):
                per_rt = avail[used] // self.job_res[j][used, None]
            self.cap[j] = per_rt.min(axis=0) if used.any() else 0
        # Spread each job's min_replicas greedily over node columns.
        self.floor = np.zeros_like(base)
        for j, job in enumerate(jobs):
            need = job.min_replicas
            for n in range(len(nodes)):
                take = min(need, self.cap[j, n])
                self.floor[j, n] = take
                need -= take

    # ---- objectives ------------------------------------------------------

    def _speedups(self, states):
        num_nodes = np.count_nonzero(states, axis=2)
        num_replicas = states.sum(axis=2)
        cols = [job.speedup_fn(num_nodes[:, j], num_replicas[:, j])
                for j, job in enumerate(self.jobs)]
        return np.stack(cols, axis=1).astype(np.float64)

    def _cluster_sizes(self, states):
        sizes = np.arange(states.shape[-1]) + 1
        return np.amax(np.where(states.any(axis=-2), sizes, 0), axis=-1)

    def evaluate(self, states):
        speedups = self._speedups(states)
        scaled = speedups * self.dominant_share * len(self.nodes)
        restarted = (states != self.base).any(axis=2)
        scaled[restarted] *= 1.0 - _RESTART_PENALTY
        return np.column_stack([-scaled.sum(axis=1),
                                self._cluster_sizes(states)])

    def cluster_utilities(self, states):
        """Average speedup/replica weighted by share of the most congested
        resource, per state."""
        num_replicas = states.sum(axis=2)
        speedups = self._speedups(states)
        active = states.sum(axis=1) > 0  # (pop, nodes)
        total = np.sum(active[:, :, None] * self.node_res, axis=1)
        alloc = num_replicas[:, :, None] * self.job_res
        with np.errstate(divide="ignore", invalid="ignore"):
            shares = np.where(alloc, alloc / total[:, None, :], 0.0)
            per_job = np.where(num_replicas, speedups / num_replicas, 0.0)
        return np.amax(np.sum(per_job[:, :, None] * shares, axis=1), axis=1)

    # ---- genetic operators ----------------------------------------------

    def crossover(self, pa, pb):
        """Job-wise single-point crossover + random cluster-size clamp."""
        n_jobs, width = pa.shape[1:]
        point = self.rng.integers(n_jobs, size=(pa.shape[0], 1, 1))
        take_a = np.arange(n_jobs)[None, :, None] < point
        ca = np.where(take_a, pa, pb)
        cb = np.where(take_a, pb, pa)
        lo, hi = np.sort(np.stack([self._cluster_sizes(pa),
                                   self._cluster_sizes(pb)]), axis=0)
        for child in (ca, cb):
            size = lo + self.rng.integers(1 << 30, size=lo.shape) \
                % (hi - lo + 1)
            mask = np.arange(width)[None, None, :] >= size[:, None, None]
            child[np.broadcast_to(mask, child.shape)] = 0
        return ca, cb

    def mutate(self, states):
        nonzero = np.count_nonzero(states, axis=2, keepdims=True)
        zero = states.shape[2] - nonzero
        prob = 1.0 / np.where(states > 0, nonzero, zero)
        hit = self.rng.random(states.shape) < prob
        rand = self.rng.integers(self.floor, self.cap + 1,
                                 size=states.shape)
        states = np.where(hit, rand, states)
        return np.maximum(states, self.floor)

    def repair(self, states):
        states = states.copy()
        # Pinned jobs keep their current allocation verbatim.
        if self.pinned:
            states[:, self.pinned] = self.base[self.pinned]
        # At most one multi-node ("distributed") job per node: in job
        # priority order, later distributed jobs are evicted from nodes
        # already claimed by an earlier one.
        distributed = np.count_nonzero(states, axis=2) > 1
        occupied = (states * distributed[:, :, None] > 0)
        states[occupied.cumsum(axis=1) > 1] = 0
        # Per-job replica cap: clamp cumulative counts over a random node
        # permutation so the trimming is not biased to high node indices.
        caps = np.array([[j.max_replicas] for j in self.jobs])
        perm = np.argsort(self.rng.random(states.shape), axis=2)
        shuffled = np.take_along_axis(states, perm, axis=2)
        shuffled = np.minimum(np.cumsum(shuffled, axis=2), caps)
        shuffled = np.diff(shuffled, axis=2, prepend=0)
        states = np.take_along_axis(shuffled, np.argsort(perm, axis=2),
                                    axis=2)
        # Node resource limits: clamp cumulative per-node usage in job
        # priority order, then convert back to replica counts.
        usage = states[:, :, :, None] * self.job_res[None, :, None, :]
        usage = np.minimum(np.cumsum(usage, axis=1), self.node_res)
        usage = np.diff(usage, axis=1, prepend=0)
        with np.errstate(divide="ignore", invalid="ignore"):
            states = np.amin(usage // self.job_res[None, :, None, :],
                             where=self.job_res[None, :, None, :] > 0,
                             initial=np.iinfo(np.int32).max, axis=-1)
        # Jobs that could not get min_replicas get nothing.
        mins = np.array([j.min_replicas for j in self.jobs])
        states[states.sum(axis=-1) < mins] = 0
        return states


# ---- NSGA-II core (numpy, self-contained) --------------------------------

def _nondominated_front(F):
    """Indices of the non-dominated solutions of F (n x 2, minimized)."""
    n = len(F)
    dominated = np.zeros(n, dtype=bool)
    for i in range(n):
        if dominated[i]:
            continue
        better_eq = (F <= F[i]).all(axis=1)
        better = (F < F[i]).any(axis=1)
        dominates_i = better_eq & better
        if dominates_i.any():
            dominated[i] = True
    return np.flatnonzero(~dominated)


def _fast_nondominated_sort(F):
    """Ranks (0 = best front) for each solution of F (minimized)."""
    n = len(F)
    rank = np.full(n, -1)
    remaining = np.arange(n)
    level = 0
    while len(remaining):
        front_local = _nondominated_front(F[remaining])
        rank[remaining[front_local]] = level
        remaining = np.delete(remaining, front_local)
        level += 1
    return rank


def _crowding_distance(F):
    n, m = F.shape
    if n <= 2:
        return np.full(n, np.inf)
    dist = np.zeros(n)
    for k in range(m):
        order = np.argsort(F[:, k], kind="stable")
        span = F[order[-1], k] - F[order[0], k]
        dist[order[0]] = dist[order[-1]] = np.inf
        if span <= 0:
            continue
        dist[order[1:-1]] += (F[order[2:], k] - F[order[:-2], k]) / span
    return dist


def _nsga2(problem, seeds, pop_size, n_gen):
    """Minimal elitist NSGA-II over integer allocation states."""
    rng = problem.rng
    seeds = problem.repair(seeds)
    # Fill the initial population with mutated copies of the seeds.
    reps = int(np.ceil(pop_size / len(seeds)))
    pop = np.concatenate([seeds] +
                         [problem.repair(problem.mutate(seeds))
                          for _ in range(reps - 1)])[:pop_size]
    if len(pop) < pop_size:
        extra = problem.repair(problem.mutate(
            pop[rng.integers(len(pop), size=pop_size - len(pop))]))
        pop = np.concatenate([pop, extra])
    F = problem.evaluate(pop)

    for _ in range(n_gen):
        rank = _fast_nondominated_sort(F)
        crowd = np.zeros(len(F))
        for r in np.unique(rank):
            sel = rank == r
            crowd[sel] = _crowding_distance(F[sel])
        # Binary tournament selection on (rank, -crowding).
        a = rng.integers(len(pop), size=pop_size)
        b = rng.integers(len(pop), size=pop_size)
        a_wins = (rank[a] < rank[b]) | ((rank[a] == rank[b]) &
                                        (crowd[a] >= crowd[b]))
        parents = np.where(a_wins, a, b)
        pa, pb = pop[parents[0::2]], pop[parents[1::2]]
        ca, cb = problem.crossover(pa, pb)
        children = np.concatenate([ca, cb])
        children = problem.repair(problem.mutate(children))
        Fc = problem.evaluate(children)
        # Elitist environmental selection from parents + children.
        allX = np.concatenate([pop, children])
        allF = np.concatenate([F, Fc])
        rank = _fast_nondominated_sort(allF)
        crowd = np.zeros(len(allF))
        for r in np.unique(rank):
            sel = rank == r
            crowd[sel] = _crowding_distance(allF[sel])
        order = np.lexsort((-crowd, rank))[:pop_size]
        pop, F = allX[order], allF[order]
    return pop, F
