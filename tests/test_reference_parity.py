"""Differential parity: run the UPSTREAM reference implementation (mounted
read-only at /root/reference) against ours on identical inputs and demand
identical outputs — split generation, checkpoint-best predicates, and the
metric serialize/reduce_sites protocol are the behaviors COINSTAC
computations depend on bit-for-bit.

These tests import the reference solely as a black-box oracle; they skip
wherever the mount is absent (e.g. on GPU boxes, which only get /root/repo).
"""
import json
import os
import sys
import types

import numpy as np
import pytest
import torch

REF_ROOT = '/root/reference'
pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF_ROOT, 'coinstac_dinunet')),
    reason='reference mount not present')


@pytest.fixture(scope='module')
def ref():
    """Import the reference package (cv2 stubbed: not in this image)."""
    if 'cv2' not in sys.modules:
        cv2 = types.ModuleType('cv2')
        cv2.createCLAHE = lambda **kw: None
        sys.modules['cv2'] = cv2
    if not hasattr(np, 'float'):  # reference uses the numpy<2 alias
        np.float = float
    sys.path.insert(0, REF_ROOT)
    import coinstac_dinunet as r
    yield r
    sys.path.remove(REF_ROOT)


def _files(n, prefix='subj'):
    return [f'{prefix}_{i:03d}.npy' for i in range(n)]


@pytest.mark.parametrize('n,ratio', [(20, (0.6, 0.2, 0.2)),
                                     (17, (0.7, 0.15, 0.15)),
                                     (9, (0.5, 0.5)),
                                     (33, (0.8, 0.1, 0.1))])
def test_ratio_split_bitwise(ref, tmp_path, n, ratio):
    from coinstac_dinunet.data import datautils as ref_du
    from coinstac_dinunet_amd.data import datautils as our_du
    a, b = tmp_path / 'ref', tmp_path / 'ours'
    a.mkdir(), b.mkdir()
    ref_du.create_ratio_split(_files(n), {'split_dir': str(a),
                                          'split_ratio': list(ratio)})
    our_du.create_ratio_split(_files(n), {'split_dir': str(b),
                                          'split_ratio': list(ratio)})
    ra, rb = sorted(os.listdir(a)), sorted(os.listdir(b))
    assert ra == rb
    for f in ra:
        sa = json.load(open(a / f))
        sb = json.load(open(b / f))
        assert sa == sb, f'{f} differs'


@pytest.mark.parametrize('n,k', [(20, 4), (18, 3), (23, 5)])
def test_kfold_split_bitwise(ref, tmp_path, n, k):
    from coinstac_dinunet.data import datautils as ref_du
    from coinstac_dinunet_amd.data import datautils as our_du
    a, b = tmp_path / 'ref', tmp_path / 'ours'
    a.mkdir(), b.mkdir()
    ref_du.create_k_fold_splits(_files(n), {'num_folds': k,
                                            'split_dir': str(a)})
    our_du.create_k_fold_splits(_files(n), {'num_folds': k,
                                            'split_dir': str(b)})
    ra, rb = sorted(os.listdir(a)), sorted(os.listdir(b))
    assert ra == rb
    for f in ra:
        assert json.load(open(a / f)) == json.load(open(b / f)), f
    # every fold partitions the full set
    for f in rb:
        s = json.load(open(b / f))
        assert sorted(s['train'] + s['validation'] + s['test']) == \
            sorted(_files(n))


@pytest.mark.parametrize('direction,scores', [
    ('maximize', [0.5, 0.6, 0.60005, 0.7, 0.69, 0.7001]),
    ('minimize', [1.0, 0.9, 0.89995, 0.5, 0.51, 0.4999]),
])
def test_performance_improved_matches(ref, direction, scores):
    from coinstac_dinunet.utils.utils import (performance_improved_ as rp,
                                              stop_training_ as rs)
    from coinstac_dinunet_amd.utils.utils import (performance_improved_ as op,
                                                  stop_training_ as os_)
    ca = {'metric_direction': direction, 'patience': 2, 'epochs': 10,
          'best_val_epoch': 0,
          'best_val_score': 0 if direction == 'maximize' else 1e11}
    cb = dict(ca)
    for ep, sc in enumerate(scores):
        assert rp(ep, sc, ca) == op(ep, sc, cb), (ep, sc)
        assert rs(ep, ca) == os_(ep, cb), (ep, sc)
        assert ca == cb  # identical cache mutations


def test_prf1a_protocol_matches(ref):
    from coinstac_dinunet.metrics import Prf1a as RefP
    from coinstac_dinunet_amd.metrics import Prf1a as OurP
    rng = np.random.RandomState(7)
    ra, oa = RefP(), OurP()
    rb, ob = RefP(), OurP()
    for m_ref, m_our in ((ra, oa), (rb, ob)):
        for _ in range(3):
            pred = torch.from_numpy(rng.randint(0, 2, 50))
            true = torch.from_numpy(rng.randint(0, 2, 50))
            m_ref.add(pred.clone(), true.clone())
            m_our.add(pred.clone(), true.clone())
    assert ra.serialize() == oa.serialize()
    assert [ra.f1, ra.accuracy, ra.precision, ra.recall] == \
           [oa.f1, oa.accuracy, oa.precision, oa.recall]
    # accumulate parity
    ra.accumulate(rb), oa.accumulate(ob)
    assert ra.serialize() == oa.serialize()
    # reduce_sites parity (mutates the instance; serialized site scores
    # -> floor values)
    site_scores = [[0.8, 0.7, 0.9], [0.6, 0.5, 0.4]]
    rr, ro = RefP(), OurP()
    rr.reduce_sites(site_scores)
    ro.reduce_sites(site_scores)
    assert rr.serialize() == ro.serialize()
    assert rr.f1 == ro.f1 and rr.accuracy == ro.accuracy


def test_averages_protocol_matches(ref):
    from coinstac_dinunet.metrics import COINNAverages as RefA
    from coinstac_dinunet_amd.metrics import COINNAverages as OurA
    ra, oa = RefA(num_averages=2), OurA(num_averages=2)
    for v, n, i in [(0.5, 4, 0), (0.25, 8, 1), (1.5, 2, 0)]:
        ra.add(v, n, index=i)
        oa.add(v, n, index=i)
    assert ra.serialize() == oa.serialize()
    np.testing.assert_allclose(np.asarray(ra.get(), dtype=float),
                               np.asarray(oa.get(), dtype=float))
    sites = [ra.serialize(), [[1.0, 2.0], [3, 5]]]
    rred, ored = RefA(num_averages=2), OurA(num_averages=2)
    rred.reduce_sites(sites)
    ored.reduce_sites(sites)
    assert rred.serialize() == ored.serialize()


def test_confusion_matrix_protocol_matches(ref):
    """The reference indexes matrix[pred][true] (metrics.py:243-249), so
    its precision()/recall() are swapped vs the standard definitions we
    use — accuracy and macro-F1 are invariant under that swap (F1 is
    symmetric in p<->r). Pin exactly that equivalence."""
    from coinstac_dinunet.metrics import ConfusionMatrix as RefC
    from coinstac_dinunet_amd.metrics import ConfusionMatrix as OurC
    rng = np.random.RandomState(11)
    rc, oc = RefC(num_classes=3), OurC(num_classes=3)
    for _ in range(3):
        pred = torch.from_numpy(rng.randint(0, 3, 60))
        true = torch.from_numpy(rng.randint(0, 3, 60))
        rc.add(pred.clone(), true.clone())
        oc.add(pred.clone(), true.clone())
    assert torch.equal(rc.matrix.long(), oc.matrix.t())  # transposed layouts
    np.testing.assert_allclose(rc.accuracy(), oc.accuracy, atol=1e-4)
    np.testing.assert_allclose(rc.f1(average=True),
                               oc.f1(average=True), atol=1e-4)
    # the reference's 'precision' is the standard recall and vice versa
    np.testing.assert_allclose(rc.precision(average=False),
                               oc.recall(average=False), atol=1e-4)
    np.testing.assert_allclose(rc.recall(average=False),
                               oc.precision(average=False), atol=1e-4)


@pytest.mark.parametrize('n,bs,shuffle,drop_last',
                         [(10, 4, True, False), (13, 4, False, False),
                          (3, 8, True, False), (10, 4, True, True)])
def test_padded_sampler_sequences_match(ref, n, bs, shuffle, drop_last):
    """Identical padded index sequences per (seed, epoch) — the property
    lock-step batch counts depend on."""
    from coinstac_dinunet.data.data import COINNPaddedDataSampler as RefS
    from coinstac_dinunet_amd.data.data import COINNPaddedDataSampler as OurS
    ds = list(range(n))
    rs = RefS(ds, bs, seed=5, shuffle=shuffle, drop_last=drop_last)
    os_ = OurS(ds, bs, seed=5, shuffle=shuffle, drop_last=drop_last)
    for epoch in range(3):
        rs.set_epoch(epoch), os_.set_epoch(epoch)
        assert list(iter(rs)) == list(iter(os_)), epoch
    assert len(rs) == len(os_)


def test_initialize_weights_bitwise(ref):
    """Same torch seed => bit-identical Kaiming init on the module types
    the reference covers (Conv2d/Linear/BatchNorm2d)."""
    from coinstac_dinunet.utils.tensorutils import (initialize_weights as ri)
    from coinstac_dinunet_amd.utils.tensorutils import (initialize_weights
                                                        as oi)

    def build():
        torch.manual_seed(42)
        return torch.nn.Sequential(
            torch.nn.Conv2d(3, 8, 3), torch.nn.BatchNorm2d(8),
            torch.nn.Flatten(), torch.nn.Linear(8, 4))

    ma, mb = build(), build()
    torch.manual_seed(99), ri(ma)
    torch.manual_seed(99), oi(mb)
    for pa, pb in zip(ma.parameters(), mb.parameters()):
        assert torch.equal(pa, pb)


def test_config_precedence_matches(ref, tmp_path):
    """Three-source hyperparameter resolution (platform input >
    {task}_args > {engine}_args > {task}_data_conf > ctor defaults,
    local.py:92-118) must cache identical values in both stacks."""
    from coinstac_dinunet import COINNLocal as RefLocal
    from coinstac_dinunet_amd import COINNLocal as OurLocal

    spec = {
        'task_id': 'tsk', 'mode': 'train', 'agg_engine': 'dSGD',
        'batch_size': 16,                       # platform input
        'tsk_args': {'epochs': 7, 'learning_rate': 0.005},
        'dSGD_args': {'local_iterations': 3},
        'tsk_data_conf': {'batch_size': 99,      # loses: already in input?
                          'data_dir': 'vols', 'num_folds': 4},
    }
    kw = dict(task_id='tsk', mode='train', batch_size=4, epochs=2,
              local_iterations=1, split_ratio=(0.8, 0.1, 0.1),
              data_dir='data', num_class=2, verbose=False)

    caches = {}
    for name, cls in (('ref', RefLocal), ('ours', OurLocal)):
        cache = {}
        cls(cache=cache, input=dict(spec), state={
            'clientId': 'local0', 'baseDirectory': str(tmp_path / name),
            'transferDirectory': str(tmp_path / name),
            'outputDirectory': str(tmp_path / name),
            'cacheDirectory': str(tmp_path / name)}, **kw)
        caches[name] = cache

    for k in ('batch_size', 'epochs', 'learning_rate', 'local_iterations',
              'data_dir', 'num_folds', 'task_id', 'mode', 'agg_engine'):
        assert caches['ref'].get(k) == caches['ours'].get(k), \
            (k, caches['ref'].get(k), caches['ours'].get(k))


def test_aucroc_protocol_matches(ref):
    from coinstac_dinunet.metrics import AUCROCMetrics as RefA
    from coinstac_dinunet_amd.metrics import AUCROCMetrics as OurA
    rng = np.random.RandomState(5)
    ra, oa = RefA(), OurA()
    for _ in range(3):
        prob = torch.from_numpy(rng.rand(40).astype(np.float32))
        true = torch.from_numpy((rng.rand(40) > 0.4).astype(np.int64))
        ra.add(prob.clone(), true.clone())
        oa.add(prob.clone(), true.clone())
    # reference exposes auc() as a method; ours as a property
    assert abs(ra.auc() - oa.auc) < 1e-6
    np.testing.assert_allclose(np.asarray(ra.serialize(), dtype=float),
                               np.asarray(oa.serialize(), dtype=float),
                               atol=1e-5)
    rr, ro = RefA(), OurA()
    rr.reduce_sites([0.7, 0.9])
    ro.reduce_sites([0.7, 0.9])
    assert abs(rr.auc() - ro.auc) < 1e-9
