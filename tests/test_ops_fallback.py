"""CPU op semantics vs independent oracles (torch.optim / numpy)."""
import numpy as np
import torch

from draco_amd.ops import fallback as fb


def test_sgd_matches_torch_after_first_step():
    torch.manual_seed(0)
    p_ref = torch.randn(100)
    p = p_ref.clone()
    ref_param = torch.nn.Parameter(p_ref.clone())
    opt = torch.optim.SGD([ref_param], lr=0.1, momentum=0.9, weight_decay=0.01)
    buf = torch.zeros_like(p)
    first = True
    for i in range(5):
        g = torch.randn(100)
        ref_param.grad = g.clone()
        opt.step()
        fb.fused_sgd_step(p, g, buf, lr=0.1, momentum=0.9, dampening=0.0,
                          weight_decay=0.01, nesterov=False, first_step=first)
        first = False
    assert torch.allclose(p, ref_param.detach(), atol=1e-6)


def test_sgd_nesterov():
    torch.manual_seed(1)
    p = torch.randn(64)
    ref_param = torch.nn.Parameter(p.clone())
    opt = torch.optim.SGD([ref_param], lr=0.05, momentum=0.8, nesterov=True)
    buf = torch.zeros_like(p)
    first = True
    for _ in range(4):
        g = torch.randn(64)
        ref_param.grad = g.clone()
        opt.step()
        fb.fused_sgd_step(p, g, buf, lr=0.05, momentum=0.8, dampening=0.0,
                          weight_decay=0.0, nesterov=True, first_step=first)
        first = False
    assert torch.allclose(p, ref_param.detach(), atol=1e-6)


def test_adam_matches_torch():
    torch.manual_seed(2)
    p = torch.randn(128)
    ref_param = torch.nn.Parameter(p.clone())
    opt = torch.optim.Adam([ref_param], lr=1e-2, betas=(0.9, 0.999), eps=1e-8)
    m = torch.zeros_like(p)
    v = torch.zeros_like(p)
    for t in range(1, 6):
        g = torch.randn(128)
        ref_param.grad = g.clone()
        opt.step()
        fb.fused_adam_step(p, g, m, v, None, step=t, lr=1e-2, beta1=0.9, beta2=0.999,
                           eps=1e-8, weight_decay=0.0, amsgrad=False)
    assert torch.allclose(p, ref_param.detach(), atol=1e-5)


def test_inject_modes():
    g = torch.randn(32)
    x = g.clone()
    fb.inject_(x, "rev_grad")
    assert torch.allclose(x, -100.0 * g)
    x = g.clone()
    fb.inject_(x, "constant")
    assert torch.all(x == -100.0)
    x = g.clone()
    fb.inject_(x, "rev_grad", cyclic=True)
    assert torch.allclose(x, g - 100.0 * g)
    x = g.clone()
    fb.inject_(x, "random")  # reference passthrough
    assert torch.equal(x, g)


def test_rows_equal_and_mean():
    x = torch.randn(6, 40)
    x[3] = x[0]
    a = torch.tensor([0, 0, 1])
    b = torch.tensor([3, 1, 2])
    eq = fb.rows_equal(x, a, b, atol=0.0)
    assert eq.tolist() == [1, 0, 0]
    out = torch.empty(40)
    fb.mean_rows(x, torch.tensor([0, 3]), out)
    assert torch.allclose(out, x[0])


def test_cyclic_ops_match_numpy():
    rng = np.random.default_rng(0)
    k, n, d = 5, 8, 64
    g = torch.tensor(rng.normal(size=(k, d)), dtype=torch.float32)
    wre = torch.tensor(rng.normal(size=k), dtype=torch.float32)
    wim = torch.tensor(rng.normal(size=k), dtype=torch.float32)
    out = torch.zeros(2, d)
    fb.cyclic_encode(g, wre, wim, out)
    ref = (g.numpy() * wre.numpy()[:, None]).sum(0)
    assert np.allclose(out[0].numpy(), ref, atol=1e-5)

    r = torch.tensor(rng.normal(size=(2 * n, d)), dtype=torch.float32)
    z = torch.tensor(rng.normal(size=d), dtype=torch.float32)
    proj = fb.cyclic_project(r, z)
    assert np.allclose(proj.numpy(), (r.numpy() @ z.numpy()), atol=1e-4)

    rows = torch.tensor([0, 3, 5])
    w = torch.tensor(rng.normal(size=3), dtype=torch.float32)
    outc = torch.zeros(d)
    fb.combine_rows(r, rows, w, outc)
    assert np.allclose(outc.numpy(), w.numpy() @ r.numpy()[rows.numpy()], atol=1e-4)
    r = r.reshape(n, 2, d)

    vre = torch.tensor(rng.normal(size=n), dtype=torch.float32)
    vim = torch.tensor(rng.normal(size=n), dtype=torch.float32)
    outd = torch.zeros(d)
    fb.cyclic_recombine(r, vre, vim, outd)
    ref = (vre.numpy() @ r.numpy()[:, 0, :]) - (vim.numpy() @ r.numpy()[:, 1, :])
    assert np.allclose(outd.numpy(), ref, atol=1e-4)


def test_segment_ops_match_bruteforce():
    rng = np.random.default_rng(1)
    P, d = 4, 50
    seg = torch.tensor([0, 7, 7, 30, 50])
    x = torch.tensor(rng.normal(size=(P, d)), dtype=torch.float32)
    z = torch.tensor(rng.normal(size=d), dtype=torch.float32)
    out = fb.segment_sqdist(x, z, seg)
    for l in range(4):
        lo, hi = int(seg[l]), int(seg[l + 1])
        ref = ((x[:, lo:hi] - z[lo:hi]) ** 2).sum(dim=1)
        assert torch.allclose(out[:, l], ref, atol=1e-4)

    w = torch.tensor(rng.normal(size=(P, 4)), dtype=torch.float32)
    outd = torch.zeros(d)
    fb.segment_weighted_mean(x, w, seg, outd)
    for l in range(4):
        lo, hi = int(seg[l]), int(seg[l + 1])
        ref = (w[:, l : l + 1] * x[:, lo:hi]).sum(dim=0)
        assert torch.allclose(outd[lo:hi], ref, atol=1e-4)

    gram = fb.segment_gram(x, seg)
    for l in range(4):
        lo, hi = int(seg[l]), int(seg[l + 1])
        ref = x[:, lo:hi] @ x[:, lo:hi].T
        assert torch.allclose(gram[l], ref, atol=1e-4)
