"""Regression tests for the round-1 advisor findings fixed in round 2."""
import numpy as np
import torch


def test_padded_sampler_epoch_advances_between_rebuilds():
    """the train loader is rebuilt each epoch; the shuffle
    order must change across epochs (seeded by seed + epoch) while
    staying identical ACROSS sites for the same epoch."""
    from coinstac_dinunet_amd.data.data import COINNDataHandle

    class _DS(torch.utils.data.Dataset):
        def __len__(self):
            return 12

        def __getitem__(self, i):
            return i

    orders = []
    cache = {'batch_size': 4, 'seed': 5, 'shuffle': True, 'cursor': 0,
             'dataloader_args': {}}
    h = COINNDataHandle(cache=cache, input={}, state={'clientId': 'a'})
    h.dataset['train'] = _DS()
    for epoch in range(3):
        seen = []
        while True:
            batch, out = h.next_iter()
            seen.extend(int(v) for v in batch)
            if out.get('mode') is not None:  # epoch ended, cursor reset
                break
        orders.append(tuple(seen))
    assert orders[0] != orders[1] or orders[1] != orders[2], orders
    # same seed+epoch on a "second site" reproduces epoch 0's order
    cache2 = {'batch_size': 4, 'seed': 5, 'shuffle': True, 'cursor': 0,
              'dataloader_args': {}}
    h2 = COINNDataHandle(cache=cache2, input={}, state={'clientId': 'b'})
    h2.dataset['train'] = _DS()
    seen2 = []
    while True:
        batch, out = h2.next_iter()
        seen2.extend(int(v) for v in batch)
        if out.get('mode') is not None:
            break
    assert tuple(seen2) == orders[0]


def test_auc_fallback_midranks_match_sklearn_on_ties():
    """the sklearn-free AUC must use midranks for tied
    probabilities."""
    from coinstac_dinunet_amd.metrics.metrics import AUCROCMetrics
    import unittest.mock as mock

    y = [0, 1, 0, 1, 1, 0, 1, 0, 0, 1]
    p = [0.3, 0.7, 0.7, 0.7, 0.9, 0.1, 0.3, 0.3, 0.5, 0.5]

    from sklearn.metrics import roc_curve, auc as _auc
    fpr, tpr, _ = roc_curve(y, p)
    expect = round(float(_auc(fpr, tpr)), 5)

    m = AUCROCMetrics()
    m.probabilities, m.labels = list(p), list(y)
    import builtins
    real_import = builtins.__import__

    def no_sklearn(name, *a, **k):
        if name.startswith('sklearn'):
            raise ImportError('forced')
        return real_import(name, *a, **k)

    with mock.patch.object(builtins, '__import__', no_sklearn):
        got = m.auc
    assert abs(got - expect) < 1e-4, (got, expect)


def test_grad_alignment_with_frozen_param():
    """extract/assign must not shift indices past a
    grad-less (frozen) parameter."""
    from coinstac_dinunet_amd.utils import tensorutils as tu

    torch.manual_seed(0)
    net = torch.nn.Sequential(torch.nn.Linear(4, 3), torch.nn.Linear(3, 2))
    net[0].weight.requires_grad_(False)  # frozen: no grad
    out = net(torch.randn(5, 4))
    out.sum().backward()

    grads = tu.extract_grads(net)
    assert len(grads) == 3  # bias0, weight1, bias1 (weight0 skipped)

    # round-trip through the learner's filtered assignment
    from coinstac_dinunet_amd.distrib.learner import COINNLearner

    class _T:
        pass

    t = _T()
    t.cache, t.input, t.state = {}, {}, {}
    t.nn = {'m': net}
    t.optimizer = {}
    t.device = {}
    learner = COINNLearner.__new__(COINNLearner)
    learner.trainer = t
    learner.device = torch.device('cpu')
    marked = [np.full_like(g, i, dtype=np.float32)
              for i, g in enumerate(grads)]
    learner._adopt_grads(marked)
    assert net[0].weight.grad is None  # untouched
    assert float(net[0].bias.grad.flatten()[0]) == 0.0
    assert float(net[1].weight.grad.flatten()[0]) == 1.0
    assert float(net[1].bias.grad.flatten()[0]) == 2.0


def test_fused_adam_buckets_by_step():
    """params whose grads first appear later must get their
    own bias-correction step count (verified via call capture)."""
    from coinstac_dinunet_amd import ops
    import unittest.mock as mock

    p1 = torch.nn.Parameter(torch.randn(4))
    p2 = torch.nn.Parameter(torch.randn(4))
    opt = ops.FusedAdam([p1, p2], lr=0.1)

    calls = []

    class _FakeC:
        @staticmethod
        def fused_adam(params, grads, m, v, lr, b1, b2, eps, wd, step):
            calls.append((len(params), step))

    with mock.patch.object(ops, 'require_native', lambda: _FakeC):
        p1.grad = torch.ones(4)
        opt.step()                      # p1 at step 1
        p2.grad = torch.ones(4)
        opt.step()                      # p1 at 2, p2 at 1: TWO buckets
    assert calls[0] == (1, 1)
    assert sorted(calls[1:]) == [(1, 1), (1, 2)], calls


def test_auc_fallback_property_sweep():
    """Randomized tie-heavy sweep: the sklearn-free AUC must track
    sklearn within rounding for any tie structure."""
    import unittest.mock as mock
    import builtins
    from sklearn.metrics import roc_curve, auc as _auc
    from coinstac_dinunet_amd.metrics.metrics import AUCROCMetrics

    rng = np.random.RandomState(7)
    real_import = builtins.__import__

    def no_sklearn(name, *a, **k):
        if name.startswith('sklearn'):
            raise ImportError('forced')
        return real_import(name, *a, **k)

    for trial in range(20):
        n = rng.randint(5, 60)
        # quantized probabilities force ties; ensure both classes present
        p = np.round(rng.rand(n), 1).tolist()
        y = rng.randint(0, 2, n)
        if y.sum() in (0, n):
            y[0] = 1 - y[0]
        y = y.tolist()
        fpr, tpr, _ = roc_curve(y, p)
        expect = float(_auc(fpr, tpr))
        m = AUCROCMetrics()
        m.probabilities, m.labels = list(p), list(y)
        with mock.patch.object(builtins, '__import__', no_sklearn):
            got = m.auc
        assert abs(got - expect) < 1e-4, (trial, got, expect)
