"""HIP kernel numerics vs plain PyTorch fp32 references (MI355X only)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from coinstac_dinunet_amd import ops
    C = ops.require_native()


@pytest.fixture(scope='module')
def dev():
    assert torch.cuda.is_available()
    return torch.device('cuda:0')


def test_fused_adam_matches_torch(dev):
    torch.manual_seed(0)
    shapes = [(64, 33), (128,), (7, 5, 3), (1025,)]
    p_ref = [torch.randn(s, device=dev) for s in shapes]
    p_hip = [p.clone() for p in p_ref]
    for p in p_ref + p_hip:
        p.requires_grad_(True)
    grads = [torch.randn(s, device=dev) for s in shapes]
    opt_ref = torch.optim.Adam(p_ref, lr=1e-2, weight_decay=0.01)
    opt_hip = ops.FusedAdam(p_hip, lr=1e-2, weight_decay=0.01)
    for step in range(5):
        for p, g in zip(p_ref, grads):
            p.grad = (g * (step + 1)).clone()
        for p, g in zip(p_hip, grads):
            p.grad = (g * (step + 1)).clone()
        opt_ref.step()
        opt_hip.step()
    for a, b in zip(p_ref, p_hip):
        torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_fused_adam_flat_matches_torch(dev):
    torch.manual_seed(1)
    n = 10007
    p_ref = torch.randn(n, device=dev, requires_grad=True)
    p_hip = p_ref.detach().clone().requires_grad_(True)
    g = torch.randn(n, device=dev)
    m = torch.zeros(n, device=dev)
    v = torch.zeros(n, device=dev)
    opt_ref = torch.optim.Adam([p_ref], lr=3e-3)
    for step in range(1, 4):
        p_ref.grad = g.clone()
        opt_ref.step()
        C.fused_adam_flat(p_hip.data, g, m, v, 3e-3, 0.9, 0.999, 1e-8, 0.0,
                          step)
    torch.testing.assert_close(p_ref, p_hip, rtol=1e-5, atol=1e-6)


def test_fused_sgd_matches_torch(dev):
    torch.manual_seed(2)
    p_ref = torch.randn(513, device=dev, requires_grad=True)
    p_hip = p_ref.detach().clone().requires_grad_(True)
    g = torch.randn(513, device=dev)
    opt_ref = torch.optim.SGD([p_ref], lr=0.1, momentum=0.9)
    opt_hip = ops.FusedSGD([p_hip], lr=0.1, momentum=0.9)
    for _ in range(4):
        p_ref.grad = g.clone()
        p_hip.grad = g.clone()
        opt_ref.step()
        opt_hip.step()
    torch.testing.assert_close(p_ref, p_hip, rtol=1e-5, atol=1e-6)


def test_lsnll_forward_backward(dev):
    torch.manual_seed(3)
    for B, Cc in [(32, 2), (128, 10), (7, 33)]:
        logits = torch.randn(B, Cc, device=dev, requires_grad=True)
        target = torch.randint(0, Cc, (B,), device=dev)
        loss = ops.cross_entropy(logits, target)
        ref_logits = logits.detach().clone().requires_grad_(True)
        ref = torch.nn.functional.cross_entropy(ref_logits, target)
        torch.testing.assert_close(loss, ref, rtol=1e-5, atol=1e-6)
        loss.backward()
        ref.backward()
        torch.testing.assert_close(logits.grad, ref_logits.grad,
                                   rtol=1e-4, atol=1e-6)


def test_lsnll_bf16(dev):
    torch.manual_seed(4)
    logits = torch.randn(64, 4, device=dev, dtype=torch.bfloat16,
                         requires_grad=True)
    target = torch.randint(0, 4, (64,), device=dev)
    loss = ops.cross_entropy(logits, target)
    ref = torch.nn.functional.cross_entropy(logits.detach().float(), target)
    torch.testing.assert_close(loss.float(), ref, rtol=1e-2, atol=1e-3)
    loss.backward()
    assert logits.grad is not None and logits.grad.dtype == torch.bfloat16


def test_argmax_rows(dev):
    x = torch.randn(257, 19, device=dev)
    torch.testing.assert_close(ops.argmax_rows(x), torch.argmax(x, dim=1))


def test_prf1a_counts(dev):
    torch.manual_seed(5)
    pred = torch.randint(0, 2, (10001,), device=dev)
    true = torch.randint(0, 2, (10001,), device=dev)
    tp, fp, tn, fn = ops.prf1a_counts(pred, true)
    pc, tc = pred.cpu(), true.cpu()
    cases = tc * 2 + pc
    assert tp == int((cases == 3).sum())
    assert fp == int((cases == 1).sum())
    assert tn == int((cases == 0).sum())
    assert fn == int((cases == 2).sum())


def test_confusion_matrix(dev):
    torch.manual_seed(6)
    K = 7
    pred = torch.randint(0, K, (5000,), device=dev)
    true = torch.randint(0, K, (5000,), device=dev)
    mat = ops.confusion_matrix(pred, true, K).cpu()
    idx = true.cpu() * K + pred.cpu()
    ref = torch.bincount(idx, minlength=K * K).reshape(K, K)
    assert torch.equal(mat, ref)


def test_pack_unpack_roundtrip(dev):
    torch.manual_seed(7)
    tensors = [torch.randn(s, device=dev) for s in [(17,), (33, 3), (257,)]]
    total = sum(t.numel() for t in tensors)
    flat = torch.empty(total, device=dev)
    C.pack_tensors(tensors, flat)
    ref = torch.cat([t.reshape(-1) for t in tensors])
    torch.testing.assert_close(flat, ref)
    outs = [torch.zeros_like(t) for t in tensors]
    C.unpack_tensors(flat, outs)
    for a, b in zip(outs, tensors):
        torch.testing.assert_close(a, b)


def test_linear_fwd_fp32(dev):
    torch.manual_seed(8)
    for M, K, N in [(64, 66, 256), (100, 17, 33), (4, 8, 2)]:
        x = torch.randn(M, K, device=dev)
        w = torch.randn(N, K, device=dev)
        b = torch.randn(N, device=dev)
        out = ops.linear(x, w, b)
        ref = torch.nn.functional.linear(x, w, b)
        torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


def test_linear_fwd_relu_and_backward(dev):
    torch.manual_seed(9)
    x = torch.randn(32, 20, device=dev, requires_grad=True)
    w = torch.randn(16, 20, device=dev, requires_grad=True)
    b = torch.randn(16, device=dev, requires_grad=True)
    out = ops.linear(x, w, b, relu=True)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    ref = torch.relu(torch.nn.functional.linear(xr, wr, br))
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)
    g = torch.randn_like(ref)
    out.backward(g)
    ref.backward(g)
    torch.testing.assert_close(x.grad, xr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w.grad, wr.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(b.grad, br.grad, rtol=1e-4, atol=1e-4)


def test_linear_bf16(dev):
    torch.manual_seed(10)
    x = torch.randn(64, 66, device=dev, dtype=torch.bfloat16)
    w = torch.randn(32, 66, device=dev, dtype=torch.bfloat16)
    out = ops.linear(x, w)
    ref = torch.nn.functional.linear(x.float(), w.float())
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)


def test_linear_bwd_kernels_vs_fp32(dev):
    """linear_dgrad/linear_wgrad MFMA kernels directly vs fp32 matmul;
    asymmetric spikes catch transposed staging."""
    C = ops.require_native()
    torch.manual_seed(21)
    for M, N, K in [(64, 64, 64), (100, 33, 17), (7, 130, 66), (1024, 256, 66)]:
        go = torch.randn(M, N, device=dev)
        w = torch.randn(N, K, device=dev)
        x = torch.randn(M, K, device=dev)
        go[min(2, M - 1), min(5, N - 1)] += 7.0
        w[min(3, N - 1), min(1, K - 1)] += 5.0
        gx = C.linear_dgrad(go, w)
        torch.testing.assert_close(gx, go @ w, rtol=1e-4, atol=1e-3)
        gw = C.linear_wgrad(go, x)
        torch.testing.assert_close(gw, go.t() @ x, rtol=1e-4, atol=1e-3)


def test_linear_bwd_kernels_bf16(dev):
    C = ops.require_native()
    torch.manual_seed(22)
    go = torch.randn(64, 32, device=dev, dtype=torch.bfloat16)
    w = torch.randn(32, 66, device=dev, dtype=torch.bfloat16)
    x = torch.randn(64, 66, device=dev, dtype=torch.bfloat16)
    gx = C.linear_dgrad(go, w)
    assert gx.dtype == torch.bfloat16
    torch.testing.assert_close(gx.float(), go.float() @ w.float(),
                               rtol=2e-2, atol=2e-2)
    gw = C.linear_wgrad(go, x)
    assert gw.dtype == torch.float32  # master-weight gradient
    torch.testing.assert_close(gw, go.float().t() @ x.float(),
                               rtol=2e-2, atol=2e-2)


def test_graft_smoke(dev):
    import __graft_entry__
    __graft_entry__.smoke()


def test_gram_schmidt_kernel(dev):
    torch.manual_seed(11)
    m = torch.randn(500, 4, device=dev)
    ref = m.clone()
    C.gram_schmidt(m, 1e-8)
    # CPU reference Gram-Schmidt
    n_cols = ref.shape[1]
    for i in range(n_cols):
        col = ref[:, i:i + 1]
        col.div_(torch.norm(col) + 1e-8)
        if i + 1 < n_cols:
            ref[:, i + 1:].sub_(col @ (col.t() @ ref[:, i + 1:]))
    torch.testing.assert_close(m, ref, rtol=1e-4, atol=1e-5)
    gram = m.t() @ m
    torch.testing.assert_close(gram, torch.eye(4, device=dev),
                               rtol=1e-3, atol=1e-3)


def test_forced_1rank_allreduce_is_identity(dev):
    """COINN_FORCE_ALLREDUCE runs the real RCCL collectives at world 1;
    an all-reduce over one rank must leave gradients unchanged."""
    import os
    import torch.distributed as tdist
    from coinstac_dinunet_amd.parallel.engine import (FlatGradBuffer,
                                                      init_distributed)
    init_distributed()
    os.environ['COINN_FORCE_ALLREDUCE'] = '1'
    try:
        net = torch.nn.Sequential(torch.nn.Linear(64, 64),
                                  torch.nn.Linear(64, 8)).to(dev)
        comm = torch.cuda.Stream()
        buf = FlatGradBuffer(net.parameters(), bucket_bytes=2048,
                             world_size=1, comm_stream=comm)
        assert buf.force_collectives and len(buf.buckets) >= 2
        x = torch.randn(16, 64, device=dev)
        buf.zero_()
        buf.begin_round(sync=False)
        net(x).sum().backward()
        ref = buf.flat.detach().clone()
        buf.zero_()
        buf.begin_round(sync=True)
        net(x).sum().backward()
        buf.finish_round()
        torch.cuda.synchronize()
        torch.testing.assert_close(buf.flat, ref)
    finally:
        os.environ.pop('COINN_FORCE_ALLREDUCE', None)
