"""Property-based ElasticSampler tests (hypothesis).

The elasticity contract (reference data.py:41-111): a deterministic
shuffle keyed on the epoch, a strided partition across replicas, and
resumability from any global sample index — so a pass interrupted at an
arbitrary point and resumed at a DIFFERENT replica count still covers
every remaining dataset index exactly once (modulo the final-batch
padding duplicates).
"""

from hypothesis import given, settings, strategies as st

from adaptdl_amd.torch.data import ElasticSampler


class _FakeDataset:
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n


def _collect(n, num_replicas, epoch, index):
    """All indices yielded across replicas for one (resumed) pass."""
    out = []
    pads = 0
    for rank in range(num_replicas):
        s = ElasticSampler(_FakeDataset(n), shuffle=True)
        s.num_replicas = num_replicas
        s.rank = rank
        s.set_epoch(epoch, index=index)
        got = list(s)
        assert len(got) == len(s)
        out.append(got)
    lens = {len(g) for g in out}
    assert len(lens) == 1, "replicas must agree on the pass length"
    flat = [i for g in out for i in g]
    return flat


@settings(max_examples=30, deadline=None)
@given(n=st.integers(min_value=4, max_value=257),
       replicas=st.integers(min_value=1, max_value=8),
       epoch=st.integers(min_value=0, max_value=3))
def test_full_pass_covers_dataset(n, replicas, epoch):
    flat = _collect(n, replicas, epoch, index=0)
    # padding may duplicate < num_replicas samples, never lose any
    assert set(flat) == set(range(n))
    assert len(flat) - n < replicas


@settings(max_examples=30, deadline=None)
@given(n=st.integers(min_value=8, max_value=200),
       r1=st.integers(min_value=1, max_value=8),
       r2=st.integers(min_value=1, max_value=8),
       frac=st.floats(min_value=0.0, max_value=0.99))
def test_rescale_resume_covers_remainder(n, r1, r2, frac):
    """Stop a pass at an arbitrary global index, restart with a
    different replica count: the union of samples seen must be the
    whole dataset, with only padding-size duplication."""
    epoch = 1
    # phase 1: run r1 replicas from index 0 and take the first `cut`
    # samples in global order (cut = what the loop consumed before the
    # rescale signal).
    cut = int(frac * n)
    s0 = ElasticSampler(_FakeDataset(n), shuffle=True)
    s0.num_replicas = 1
    s0.rank = 0
    s0.set_epoch(epoch, index=0)
    order = list(s0)          # the canonical shuffled order
    seen_before = set(order[:cut])
    # phase 2: resume at global index `cut` with r2 replicas
    flat = _collect(n, r2, epoch, index=cut)
    assert seen_before | set(flat) == set(range(n))
    # the resumed pass yields exactly the not-yet-seen remainder
    # (plus bounded padding duplicates)
    assert len(flat) - (n - cut) < r2
    assert set(order[cut:]) <= set(flat)
