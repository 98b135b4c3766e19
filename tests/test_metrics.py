import pytest
import torch

from coinstac_dinunet_amd.metrics import (AUCROCMetrics, COINNAverages,
                                          ConfusionMatrix, Prf1a)


def test_prf1a_counts_hand_computed():
    m = Prf1a()
    pred = torch.tensor([1, 0, 1, 1, 0, 0])
    true = torch.tensor([1, 0, 0, 1, 1, 0])
    m.add(pred, true)
    assert (m.tp, m.fp, m.tn, m.fn) == (2, 1, 2, 1)
    assert m.precision == pytest.approx(2 / 3, abs=1e-4)
    assert m.recall == pytest.approx(2 / 3, abs=1e-4)
    assert m.accuracy == pytest.approx(4 / 6, abs=1e-4)
    assert m.f1 == pytest.approx(2 / 3, abs=1e-4)


def test_prf1a_accumulate_and_serialize():
    a, b = Prf1a(), Prf1a()
    a.add(torch.tensor([1, 1]), torch.tensor([1, 0]))
    b.add(torch.tensor([0, 1]), torch.tensor([0, 1]))
    a.accumulate(b)
    assert a.tp == 2 and a.fp == 1 and a.tn == 1
    acc, prec, rec = a.serialize()
    assert acc == a.accuracy and prec == a.precision and rec == a.recall


def test_prf1a_reduce_sites_is_unweighted_mean():
    m = Prf1a()
    m.reduce_sites([[0.9, 0.8, 0.7], [0.5, 0.4, 0.3]])
    assert m.accuracy == pytest.approx(0.7, abs=1e-4)
    assert m.precision == pytest.approx(0.6, abs=1e-4)
    assert m.recall == pytest.approx(0.5, abs=1e-4)


def test_averages_weighted_reduce():
    a = COINNAverages(num_averages=1)
    a.add(2.0, n=4)
    assert a.get() == [2.0]
    # reduce_sites SUMS values+counts => data-weighted mean
    s1 = [[8.0], [4]]   # mean 2.0 over 4 samples
    s2 = [[2.0], [1]]   # mean 2.0 over 1 sample... use different: 6.0/1
    s2 = [[6.0], [1]]
    r = COINNAverages(num_averages=1)
    r.reduce_sites([s1, s2])
    assert r.get() == [pytest.approx((8 + 6) / 5, abs=1e-4)]


def test_confusion_matrix():
    m = ConfusionMatrix(num_classes=3)
    m.add(torch.tensor([0, 1, 2, 2]), torch.tensor([0, 1, 1, 2]))
    expected = torch.zeros(3, 3, dtype=torch.long)
    expected[0, 0] = 1
    expected[1, 1] = 1
    expected[1, 2] = 1
    expected[2, 2] = 1
    assert torch.equal(m.matrix, expected)
    assert m.accuracy == pytest.approx(0.75, abs=1e-4)
    ser = m.serialize()
    assert len(ser) == 3 and len(ser[1]) == 3


def test_auc_matches_sklearn():
    m = AUCROCMetrics()
    probs = torch.tensor([0.9, 0.8, 0.3, 0.2, 0.6])
    labels = torch.tensor([1, 1, 0, 0, 1])
    m.add(probs, labels)
    from sklearn.metrics import roc_auc_score
    assert m.auc == pytest.approx(
        roc_auc_score(labels.numpy(), probs.numpy()), abs=1e-4)


def test_auc_reduce_sites_mean():
    m = AUCROCMetrics()
    m.reduce_sites([[0.8], [0.6]])
    assert m.auc == pytest.approx(0.7, abs=1e-4)


def test_dice_loss_binary():
    from coinstac_dinunet_amd.metrics import dice_loss_binary
    perfect = torch.tensor([1.0, 0.0, 1.0, 1.0])
    target = torch.tensor([1.0, 0.0, 1.0, 1.0])
    assert float(dice_loss_binary(perfect, target)) < 1e-4
    worst = torch.tensor([0.0, 1.0, 0.0, 0.0])
    assert float(dice_loss_binary(worst, target)) > 0.9
