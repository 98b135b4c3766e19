"""Property-based invariants (hypothesis): the lock-step guarantees the
phase machine depends on must hold for ARBITRARY site/data geometry,
not just the fixtures the example tests use."""
import json

import numpy as np
from hypothesis import given, settings, strategies as st

from coinstac_dinunet_amd.data import datautils
from coinstac_dinunet_amd.data.data import COINNPaddedDataSampler


@settings(max_examples=60, deadline=None)
@given(n=st.integers(1, 200), bs=st.integers(1, 32),
       seed=st.integers(0, 2**31 - 1), shuffle=st.booleans(),
       epoch=st.integers(0, 5))
def test_padded_sampler_covers_and_pads(n, bs, seed, shuffle, epoch):
    ds = list(range(n))
    s = COINNPaddedDataSampler(ds, bs, seed=seed, shuffle=shuffle)
    s.set_epoch(epoch)
    idx = list(iter(s))
    assert len(idx) == len(s) == -(-n // bs) * bs  # ceil to batch multiple
    assert set(idx) == set(range(n))               # every sample appears
    assert all(0 <= i < n for i in idx)


@settings(max_examples=40, deadline=None)
@given(sizes=st.lists(st.integers(1, 100), min_size=2, max_size=6),
       bs=st.integers(1, 16), seed=st.integers(0, 10_000))
def test_lockstep_total_size_equalizes_batch_counts(sizes, bs, seed):
    """With total_size = max ceil-padded site size, EVERY site yields the
    same number of equal-size batches — the no-starvation guarantee."""
    total = max(-(-n // bs) * bs for n in sizes)
    counts = set()
    for n in sizes:
        s = COINNPaddedDataSampler(list(range(n)), bs, seed=seed,
                                   shuffle=True, total_size=total)
        idx = list(iter(s))
        assert len(idx) == total
        assert set(idx) <= set(range(n))
        counts.add(len(idx) // bs)
    assert len(counts) == 1


@settings(max_examples=40, deadline=None)
@given(n=st.integers(2, 120), k=st.integers(2, 8))
def test_kfold_partitions(tmp_path_factory, n, k):
    if k > n:
        return
    d = tmp_path_factory.mktemp('splits')
    files = [f'f{i:03d}' for i in range(n)]
    datautils.create_k_fold_splits(list(files),
                                   {'num_folds': k, 'split_dir': str(d)})
    import os
    names = sorted(os.listdir(d))
    assert len(names) == k
    test_sets = []
    for f in names:
        s = json.load(open(os.path.join(d, f)))
        assert sorted(s['train'] + s['validation'] + s['test']) == \
            sorted(files)
        assert len(s['test']) > 0 and len(s['validation']) > 0
        test_sets.append(frozenset(s['test']))
    # the k test parts partition the whole file set
    assert len(frozenset().union(*test_sets)) == n
    assert sum(len(t) for t in test_sets) == n


@settings(max_examples=40, deadline=None)
@given(n=st.integers(3, 120),
       ratio=st.sampled_from([(0.6, 0.2, 0.2), (0.7, 0.15, 0.15),
                              (0.8, 0.2), (0.5, 0.25, 0.25)]))
def test_ratio_split_partitions(tmp_path_factory, n, ratio):
    d = tmp_path_factory.mktemp('rsplit')
    files = [f'f{i:03d}' for i in range(n)]
    datautils.create_ratio_split(list(files), {'split_dir': str(d),
                                               'split_ratio': list(ratio)})
    import os
    s = json.load(open(os.path.join(d, os.listdir(d)[0])))
    got = s.get('train', []) + s.get('validation', []) + s.get('test', [])
    assert sorted(got) == sorted(files)  # nothing lost, nothing duplicated


@settings(max_examples=40, deadline=None)
@given(labels=st.lists(st.tuples(st.integers(0, 1), st.integers(0, 1)),
                       min_size=1, max_size=200),
       n_chunks=st.integers(1, 5))
def test_prf1a_chunked_accumulation_equals_whole(labels, n_chunks):
    """Streaming invariance: metric(all batches) == accumulate(chunks)."""
    import torch
    from coinstac_dinunet_amd.metrics import Prf1a
    pred = torch.tensor([p for p, _ in labels])
    true = torch.tensor([t for _, t in labels])
    whole = Prf1a()
    whole.add(pred, true)
    acc = Prf1a()
    for chunk in np.array_split(np.arange(len(labels)), n_chunks):
        if len(chunk) == 0:
            continue
        m = Prf1a()
        m.add(pred[chunk], true[chunk])
        acc.accumulate(m)
    assert whole.serialize() == acc.serialize()
    assert whole.f1 == acc.f1


@settings(max_examples=40, deadline=None)
@given(vals=st.lists(st.tuples(
    st.floats(-100, 100, allow_nan=False), st.integers(1, 50)),
    min_size=1, max_size=60))
def test_averages_weighted_mean_property(vals):
    """add(v, n) must produce the exact weighted mean."""
    from coinstac_dinunet_amd.metrics import COINNAverages
    a = COINNAverages(num_averages=1)
    for v, n in vals:
        a.add(v, n)
    expect = sum(v * n for v, n in vals) / sum(n for _, n in vals)
    got = np.asarray(a.average, dtype=float).reshape(-1)[0]
    assert abs(got - round(expect, 5)) < 1e-4


@settings(max_examples=30, deadline=None)
@given(shapes=st.lists(
    st.tuples(st.integers(1, 40), st.integers(0, 12)),  # (rows, cols|0=1D)
    min_size=1, max_size=10),
    bucket_kb=st.integers(1, 64))
def test_flat_grad_buffer_partitions_any_model(shapes, bucket_kb):
    """For ANY parameter geometry: views alias grads 1:1, buckets
    partition the arena exactly, and a filled arena round-trips."""
    import torch
    from coinstac_dinunet_amd.parallel.engine import FlatGradBuffer
    params = [torch.nn.Parameter(torch.randn(r, c) if c else torch.randn(r))
              for r, c in shapes]
    buf = FlatGradBuffer(params, bucket_bytes=bucket_kb * 1024, world_size=1)
    total = sum(p.numel() for p in params)
    assert buf.flat.numel() == total
    # buckets partition [0, total)
    spans = sorted((s, s + n) for s, n, _ in buf.buckets)
    assert spans[0][0] == 0 and spans[-1][1] == total
    for (a0, a1), (b0, b1) in zip(spans, spans[1:]):
        assert a1 == b0
    # every param's grad is a view into the arena
    for p in params:
        assert p.grad is not None
        assert p.grad.data_ptr() >= buf.flat.data_ptr()
        p.grad.fill_(3.0)
    assert torch.all(buf.flat == 3.0)
    buf.zero_()
    assert torch.all(buf.flat == 0.0)
    buf.remove_hooks()


@settings(max_examples=25, deadline=None)
@given(n=st.integers(4, 40), m=st.integers(4, 40),
       true_rank=st.integers(1, 4), seed=st.integers(0, 9999))
def test_power_iteration_recovers_low_rank(n, m, true_rank, seed):
    """On an exactly rank-r product, rank-r extraction must reconstruct
    it (to float tolerance) — the guarantee rankDAD's gradient fidelity
    rests on."""
    import torch
    from coinstac_dinunet_amd.distrib.rankdad import power_iteration_BC
    g = torch.Generator().manual_seed(seed)
    k = true_rank
    B = torch.randn(n, k, generator=g) @ torch.randn(k, 17, generator=g)
    C = torch.randn(m, k, generator=g) @ torch.randn(k, 17, generator=g)
    gen = torch.Generator().manual_seed(seed + 1)
    bf, cf = power_iteration_BC(B, C, rank=k + 2, numiterations=30,
                                tol=1e-6, generator=gen)
    G = B @ C.t()
    err = torch.norm(bf @ cf.t() - G) / max(torch.norm(G), 1e-8)
    assert err < 5e-2, float(err)


@settings(max_examples=30, deadline=None)
@given(items=st.lists(
    st.one_of(st.none(), st.booleans(),
              st.integers(1, 5)),  # falsy values get dropped
    min_size=0, max_size=12))
def test_safe_collate_drops_falsy(items):
    """Corrupt-sample tolerance: falsy items never reach default_collate."""
    import torch
    from coinstac_dinunet_amd.data.data import safe_collate
    batch = [{'x': torch.tensor([float(v)])} if v else v for v in items]
    kept = [b for b in batch if b]
    if not kept:
        return  # nothing to collate either way
    out = safe_collate(batch)
    assert out['x'].shape[0] == len(kept)


@settings(max_examples=30, deadline=None)
@given(depth=st.integers(0, 3), seed=st.integers(0, 999))
def test_save_cache_always_json_serializable(tmp_path_factory, depth, seed):
    """save_cache must never crash: arbitrary nests of tensors/arrays/
    callables stringify instead of raising."""
    import json
    import os
    import torch
    from coinstac_dinunet_amd import utils
    rng = np.random.RandomState(seed)

    def make(d):
        if d == 0:
            opts = [1, 'a', None, 3.5, torch.randn(2), np.arange(3), len]
            return opts[rng.randint(len(opts))]
        return {f'k{i}': make(d - 1) for i in range(2)}

    d = tmp_path_factory.mktemp('cache')
    cache = {'nested': make(depth), 'log_dir': str(d)}
    utils.save_cache(cache, str(d))
    json.load(open(os.path.join(str(d), 'logs.json')))  # parses
