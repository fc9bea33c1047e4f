"""Property-based GradSyncEngine bucket-assembly tests (hypothesis).

The engine owns the gradient memory: every parameter's ``.grad`` must be
a view into exactly one flat per-(group, dtype) bucket, laid out so
autograd's in-place accumulation writes the flat buffer directly (no
copies), including channels_last parameters whose natural gradient
layout is NHWC-strided.  These invariants carry the fused statistics,
the RCCL all-reduce, and the fused optimizers — a silent aliasing or
layout break would corrupt training, so they are pinned down over
arbitrary parameter shapes, group splits, and bucket caps.
"""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st


class _Owner:
    """Statistic-callback sink (host-side no-op)."""

    def _local_stat(self, bucket, microbatch):
        pass

    def _accum_stat(self, bucket):
        pass

    def _total_stat(self, bucket, scale):
        if scale != 1.0:
            bucket.flat.mul_(scale)

    def _on_accum_done(self):
        pass

    def _on_sync_done(self):
        pass


def _make_engine(param_groups, cap_mb):
    from adaptdl_amd.torch._engine import GradSyncEngine
    return GradSyncEngine(param_groups, owner=_Owner(),
                          bucket_cap_mb=cap_mb)


_shapes = st.lists(
    st.lists(st.integers(min_value=1, max_value=5),
             min_size=1, max_size=4),
    min_size=1, max_size=8)


@settings(max_examples=30, deadline=None)
@given(shapes=_shapes,
       n_groups=st.integers(min_value=1, max_value=3),
       cap_kb=st.sampled_from([1, 4, 1024]),
       channels_last=st.booleans(),
       data=st.data())
def test_bucket_partition_and_grad_views(shapes, n_groups, cap_kb,
                                         channels_last, data):
    params = []
    for shape in shapes:
        p = torch.nn.Parameter(torch.randn(*shape))
        if channels_last and p.dim() == 4:
            p.data = p.data.to(memory_format=torch.channels_last)
        params.append(p)
    groups = [[] for _ in range(n_groups)]
    for p in params:
        groups[data.draw(st.integers(0, n_groups - 1))].append(p)

    engine = _make_engine(groups, cap_kb / 1024.0)
    try:
        # 1. Exact partition: every requires-grad param appears in
        #    exactly one bucket segment, with matching numel, and each
        #    bucket's segments tile its flat buffer exactly.
        seen = set()
        for bucket in engine.buckets:
            offset = 0
            for p, off, n in bucket.segments:
                assert id(p) not in seen
                seen.add(id(p))
                assert off == offset
                assert n == p.numel()
                offset += n
            assert offset == bucket.flat.numel()
            assert all(gp is bucket
                       for (p, _, _) in bucket.segments
                       for gp in [engine._param_to_bucket[p]])
        assert seen == {id(p) for p in params}

        # 2. Soft cap: every bucket except a group's last may exceed the
        #    cap only because of its final parameter.
        cap_elems = cap_kb * 1024 // 4  # fp32
        for bucket in engine.buckets:
            if len(bucket.segments) > 1:
                _, _, last_n = bucket.segments[-1]
                assert bucket.flat.numel() - last_n < cap_elems

        # 3. Grad views alias the flat buffer with the parameter's own
        #    memory layout (channels_last params get NHWC-strided views).
        for bucket in engine.buckets:
            for p, off, n in bucket.segments:
                assert p.grad is not None
                assert p.grad.shape == p.shape
                assert p.grad.data_ptr() == \
                    bucket.flat[off:off + n].data_ptr()
                if p.dim() == 4:
                    assert p.grad.is_contiguous(
                        memory_format=torch.channels_last) == \
                        p.is_contiguous(
                            memory_format=torch.channels_last) or \
                        p.is_contiguous()

        # 4. Autograd accumulation writes the flat buffer in place:
        #    backward of sum(2*p) puts 2s exactly in [off, off+n).
        target = params[0]
        (2.0 * target).sum().backward()
        bucket = engine._param_to_bucket[target]
        off = next(o for (p, o, n) in bucket.segments if p is target)
        n = target.numel()
        flat = bucket.flat.detach()
        assert torch.all(flat[off:off + n] == 2.0)
        assert flat.abs().sum() == 2.0 * n  # nothing else touched

        # 5. zero_grad clears every bucket and resets counters.
        engine.accum_count = 3
        engine.zero_grad()
        for b in engine.buckets:
            assert torch.all(b.flat == 0)
            assert b.ready == 0 and b.work is None
        assert engine.accum_count == 0
        assert torch.all(engine.stats == 0)
    finally:
        engine.detach()


@settings(max_examples=15, deadline=None)
@given(shapes=_shapes, cap_kb=st.sampled_from([1, 64]))
def test_backward_flushes_every_bucket_once(shapes, cap_kb):
    """A full backward marks every bucket ready exactly once and the
    end-of-backward callback resets the ready counts."""
    params = [torch.nn.Parameter(torch.randn(*s)) for s in shapes]
    engine = _make_engine([params], cap_kb / 1024.0)
    try:
        loss = sum((p * p).sum() for p in params)
        loss.backward()
        # After the end-of-backward callback: counters reset, one
        # accumulated microbatch recorded (require_sync=True path).
        assert engine.accum_count == 1
        for b in engine.buckets:
            assert b.ready == 0
        # Gradients are 2p, element order preserved segment-by-segment.
        for b in engine.buckets:
            for p, off, n in b.segments:
                np.testing.assert_allclose(
                    p.grad.detach().numpy(),
                    2 * p.detach().numpy(), rtol=1e-5)
    finally:
        engine.detach()
