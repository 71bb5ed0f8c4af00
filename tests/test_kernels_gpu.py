"""HIP kernel numerics vs the CPU torch oracle (fp32 reference), on a real MI355X."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from draco_amd import ops
    from draco_amd.ops import fallback as fb
    from draco_amd.ops.native import available

    assert available(), "HIP extension must be built and importable on the GPU box"

DEV = "cuda:0"


def _r(*shape, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g, dtype=torch.float32)


def test_sgd_kernel_matches_oracle():
    d = 4096 + 64
    p_cpu, g_cpu = _r(d, seed=1), _r(d, seed=2)
    buf_cpu = torch.zeros(d)
    p_gpu, g_gpu, buf_gpu = p_cpu.to(DEV), g_cpu.to(DEV), buf_cpu.to(DEV)
    for first in (True, False):
        fb.fused_sgd_step(p_cpu, g_cpu, buf_cpu, lr=0.1, momentum=0.9, dampening=0.1,
                          weight_decay=0.01, nesterov=False, first_step=first)
        ops.fused_sgd_step(p_gpu, g_gpu, buf_gpu, lr=0.1, momentum=0.9, dampening=0.1,
                          weight_decay=0.01, nesterov=False, first_step=first)
    assert torch.allclose(p_cpu, p_gpu.cpu(), atol=1e-6)
    assert torch.allclose(buf_cpu, buf_gpu.cpu(), atol=1e-6)


def test_sgd_nesterov_kernel():
    d = 1024
    p_cpu, g_cpu = _r(d, seed=3), _r(d, seed=4)
    buf_cpu = torch.zeros(d)
    p_gpu, g_gpu, buf_gpu = p_cpu.to(DEV), g_cpu.to(DEV), buf_cpu.to(DEV)
    for first in (True, False, False):
        fb.fused_sgd_step(p_cpu, g_cpu, buf_cpu, lr=0.05, momentum=0.8, dampening=0.0,
                          weight_decay=0.0, nesterov=True, first_step=first)
        ops.fused_sgd_step(p_gpu, g_gpu, buf_gpu, lr=0.05, momentum=0.8, dampening=0.0,
                          weight_decay=0.0, nesterov=True, first_step=first)
    assert torch.allclose(p_cpu, p_gpu.cpu(), atol=1e-6)


def test_adam_kernel_matches_oracle():
    d = 2048
    p_cpu, g_cpu = _r(d, seed=5), _r(d, seed=6)
    m_cpu, v_cpu, x_cpu = torch.zeros(d), torch.zeros(d), torch.zeros(d)
    p_gpu, g_gpu = p_cpu.to(DEV), g_cpu.to(DEV)
    m_gpu, v_gpu, x_gpu = m_cpu.to(DEV), v_cpu.to(DEV), x_cpu.to(DEV)
    for t in range(1, 4):
        fb.fused_adam_step(p_cpu, g_cpu, m_cpu, v_cpu, x_cpu, step=t, lr=1e-2, beta1=0.9,
                           beta2=0.999, eps=1e-8, weight_decay=0.01, amsgrad=True)
        ops.fused_adam_step(p_gpu, g_gpu, m_gpu, v_gpu, x_gpu, step=t, lr=1e-2, beta1=0.9,
                            beta2=0.999, eps=1e-8, weight_decay=0.01, amsgrad=True)
    assert torch.allclose(p_cpu, p_gpu.cpu(), atol=1e-5)


def test_inject_kernel():
    d = 512
    g = _r(d, seed=7)
    for mode, cyc in [("rev_grad", False), ("rev_grad", True), ("constant", False), ("constant", True)]:
        a = g.clone()
        b = g.clone().to(DEV)
        fb.inject_(a, mode, cyc)
        ops.inject_(b, mode, cyc)
        assert torch.allclose(a, b.cpu(), atol=1e-5), (mode, cyc)


def test_rows_equal_kernel():
    m, d = 8, 4096
    x = _r(m, d, seed=8)
    x[5] = x[1]
    x[6] = x[1]
    x[6, -1] += 1e-3  # near-equal but not equal
    a = torch.tensor([1, 1, 1, 0])
    b = torch.tensor([5, 6, 2, 3])
    ref = fb.rows_equal(x, a, b, 0.0)
    got = ops.rows_equal(x.to(DEV), a.to(DEV), b.to(DEV), 0.0)
    assert torch.equal(ref, got.cpu())
    # atol mode
    ref = fb.rows_equal(x, a, b, 1e-2)
    got = ops.rows_equal(x.to(DEV), a.to(DEV), b.to(DEV), 1e-2)
    assert torch.equal(ref, got.cpu())


def test_mean_sum_rows_kernel():
    m, d = 6, 8192
    x = _r(m, d, seed=9)
    idx = torch.tensor([0, 2, 5])
    out_ref = torch.empty(d)
    fb.mean_rows(x, idx, out_ref)
    out_gpu = torch.empty(d, device=DEV)
    ops.mean_rows(x.to(DEV), idx.to(DEV), out_gpu)
    assert torch.allclose(out_ref, out_gpu.cpu(), atol=1e-5)
    fb.sum_rows(x, out_ref)
    ops.sum_rows(x.to(DEV), out_gpu)
    assert torch.allclose(out_ref, out_gpu.cpu(), atol=1e-4)


def test_cyclic_kernels():
    k, n, d = 5, 8, 65536
    g = _r(k, d, seed=10)
    wre, wim = _r(k, seed=11), _r(k, seed=12)
    out_ref = torch.zeros(2, d)
    fb.cyclic_encode(g, wre, wim, out_ref)
    out_gpu = torch.zeros(2, d, device=DEV)
    ops.cyclic_encode(g.to(DEV), wre.to(DEV), wim.to(DEV), out_gpu)
    assert torch.allclose(out_ref, out_gpu.cpu(), atol=1e-4)

    r = _r(n, 2, d, seed=13)
    z = _r(d, seed=14)
    r2 = r.view(2 * n, d)
    proj_ref = fb.cyclic_project(r2, z)
    proj_gpu = ops.cyclic_project(r2.to(DEV).contiguous(), z.to(DEV))
    assert torch.allclose(proj_ref, proj_gpu.cpu(), rtol=1e-4, atol=1e-2)

    rows = torch.tensor([1, 4, 7, 10])
    wr = _r(4, seed=21)
    outc_ref = torch.zeros(d)
    fb.combine_rows(r2, rows, wr, outc_ref)
    outc_gpu = torch.zeros(d, device=DEV)
    ops.combine_rows(r2.to(DEV).contiguous(), rows.to(DEV), wr.to(DEV), outc_gpu)
    assert torch.allclose(outc_ref, outc_gpu.cpu(), atol=1e-3)

    vre, vim = _r(n, seed=15), _r(n, seed=16)
    out_ref = torch.zeros(d)
    fb.cyclic_recombine(r, vre, vim, out_ref)
    out_gpu = torch.zeros(d, device=DEV)
    ops.cyclic_recombine(r.to(DEV).contiguous(), vre.to(DEV), vim.to(DEV), out_gpu)
    assert torch.allclose(out_ref, out_gpu.cpu(), atol=1e-3)


def test_segment_kernels():
    P, d = 7, 10000
    seg = torch.tensor([0, 1000, 1000, 4096, 9000, 10000])
    x = _r(P, d, seed=17)
    z = _r(d, seed=18)
    ref = fb.segment_sqdist(x, z, seg)
    got = ops.segment_sqdist(x.to(DEV), z.to(DEV), seg.to(DEV))
    assert torch.allclose(ref, got.cpu(), rtol=1e-4, atol=1e-2)

    w = _r(P, 5, seed=19)
    out_ref = torch.zeros(d)
    fb.segment_weighted_mean(x, w, seg, out_ref)
    out_gpu = torch.zeros(d, device=DEV)
    ops.segment_weighted_mean(x.to(DEV), w.to(DEV), seg.to(DEV), out_gpu)
    assert torch.allclose(out_ref, out_gpu.cpu(), atol=1e-4)

    gram_ref = fb.segment_gram(x, seg)
    gram_gpu = ops.segment_gram(x.to(DEV), seg.to(DEV))
    assert torch.allclose(gram_ref, gram_gpu.cpu(), rtol=1e-4, atol=1e-2)


def test_segment_absmax_kernels_match_oracle():
    torch.manual_seed(11)
    rows, d = 9, 8192
    seg = torch.tensor([0, 3, 700, 701, 4096, 8000], dtype=torch.int64)
    x = _r(rows, d, seed=12) * torch.logspace(-3, 3, d)  # wide dynamic range
    ref = fb.segment_absmax(x, seg)
    got = ops.segment_absmax(x.to(DEV), seg).cpu()
    assert torch.allclose(ref, got, atol=0, rtol=0), float((ref - got).abs().max())

    a = torch.tensor([0, 1, 4, 7], dtype=torch.int64)
    b = torch.tensor([2, 3, 5, 8], dtype=torch.int64)
    ref = fb.segment_pair_maxdiff(x, a, b, seg)
    got = ops.segment_pair_maxdiff(x.to(DEV), a, b, seg).cpu()
    assert torch.allclose(ref, got, atol=1e-6), float((ref - got).abs().max())


def test_guarded_sgd_skip_is_noop():
    d = 2048
    p = _r(d, seed=20).to(DEV)
    g = _r(d, seed=21).to(DEV)
    buf = torch.zeros(d, device=DEV)
    p0 = p.clone()
    bad = torch.tensor(False, device=DEV)
    ops.fused_sgd_step(p, g, buf, lr=0.1, momentum=0.9, dampening=0.0,
                       weight_decay=0.0, nesterov=False, first_step=True, guard=bad)
    assert torch.equal(p, p0) and torch.equal(buf, torch.zeros_like(buf))
    ok = torch.tensor(True, device=DEV)
    ops.fused_sgd_step(p, g, buf, lr=0.1, momentum=0.9, dampening=0.0,
                       weight_decay=0.0, nesterov=False, first_step=True, guard=ok)
    assert not torch.equal(p, p0), "guarded-true update must apply"
