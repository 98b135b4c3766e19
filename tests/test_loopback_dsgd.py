"""End-to-end lock-step dSGD over the loopback transport: 2 CPU sites.

This is BASELINE.json config 1 (FreeSurfer-style MLP, 2 CPU sites via
COINNLocal/COINNRemote) — the reference's whole protocol with no GPU.
"""
import os

import numpy as np
import torch

from coinstac_dinunet_amd import COINNLocal, COINNRemote
from coinstac_dinunet_amd.config.keys import Key, Mode
from coinstac_dinunet_amd.simulator import LoopbackCluster

from computations import TabularDataset, TabularTrainer, make_site_data


def _make_cluster(tmp_path, n_sites=2, n_samples=20, epochs=2, **local_kw):
    cluster = LoopbackCluster(
        str(tmp_path), n_sites=n_sites,
        site_data=lambda s: make_site_data(
            s.as_dict(), n_samples=n_samples, seed=int(s.clientId[-1])))

    kw = dict(task_id='tab', mode=Mode.TRAIN, batch_size=4, epochs=epochs,
              validation_epochs=1, local_iterations=1,
              split_ratio=(0.6, 0.2, 0.2), data_dir='data', num_class=2,
              seed_all=True, patience=epochs, verbose=False)
    kw.update(local_kw)

    def make_local(cache, input, state):
        return COINNLocal(cache=cache, input=input, state=state, **kw)

    def make_remote(cache, input, state):
        return COINNRemote(cache=cache, input=input, state=state)

    return cluster, make_local, make_remote


def test_dsgd_two_sites_full_protocol(tmp_path):
    cluster, make_local, make_remote = _make_cluster(tmp_path)
    success, out = cluster.run(make_local, make_remote, TabularTrainer,
                               dataset_cls=TabularDataset, max_rounds=400)
    assert success, f'protocol did not converge in {cluster.rounds} rounds'
    assert len(cluster.remote_cache[Key.GLOBAL_TEST_SERIALIZABLE]) == 1
    assert Key.GLOBAL_TEST_METRICS in cluster.remote_cache
    for site in cluster.sites:
        zips = [f for f in os.listdir(site.outputDirectory)
                if f.endswith('.zip')]
        assert zips, 'results zip missing on a site'


def test_dsgd_kfold_checkpointing(tmp_path):
    """3-fold cross validation: every fold trains, tests, checkpoints."""
    cluster, make_local, make_remote = _make_cluster(
        tmp_path, n_samples=18, epochs=1, split_ratio=None, num_folds=3)
    success, _ = cluster.run(make_local, make_remote, TabularTrainer,
                             dataset_cls=TabularDataset, max_rounds=600)
    assert success
    assert len(cluster.remote_cache[Key.GLOBAL_TEST_SERIALIZABLE]) == 3
    # per-fold best/latest checkpoints on every site
    for i, site in enumerate(cluster.sites):
        for fold in range(3):
            fold_dir = os.path.join(site.outputDirectory, 'tab', f'fold_{fold}')
            names = os.listdir(fold_dir)
            assert f'latest.tab-{fold}.pt' in names, (site.clientId, fold, names)


def test_dsgd_sites_stay_in_sync(tmp_path):
    """Same seeded init + same averaged gradient => weights identical
    across sites after training (lock-step dSGD equivalence)."""
    cluster, make_local, make_remote = _make_cluster(tmp_path, epochs=1)
    success, _ = cluster.run(make_local, make_remote, TabularTrainer,
                             dataset_cls=TabularDataset, max_rounds=400)
    assert success
    m0 = cluster.site_caches[0]['nn']['net']
    m1 = cluster.site_caches[1]['nn']['net']
    for p0, p1 in zip(m0.parameters(), m1.parameters()):
        assert torch.allclose(p0, p1, atol=1e-6), \
            'dSGD sites diverged despite averaged gradients'


def test_dsgd_average_equals_manual_mean(tmp_path):
    """The published average equals the hand-computed mean of the sites'
    shipped gradients (functional parity anchor: reducer.py:25-32)."""
    from coinstac_dinunet_amd.utils.tensorutils import load_arrays
    cluster, make_local, make_remote = _make_cluster(tmp_path, epochs=1)
    success, _ = cluster.run(make_local, make_remote, TabularTrainer,
                             dataset_cls=TabularDataset, max_rounds=400)
    assert success
    site_grads = []
    for site in cluster.sites:
        p = os.path.join(cluster.remote_state.baseDirectory, site.clientId,
                         'grads.npy')
        assert os.path.exists(p)
        site_grads.append(load_arrays(p))
    avg_path = os.path.join(cluster.sites[0].baseDirectory, 'avg_grads.npy')
    assert os.path.exists(avg_path)
    avg = load_arrays(avg_path)
    for i in range(len(avg)):
        manual = (np.asarray(site_grads[0][i], dtype=np.float64) +
                  np.asarray(site_grads[1][i], dtype=np.float64)) / 2
        np.testing.assert_allclose(np.asarray(avg[i], dtype=np.float64),
                                   manual, rtol=1e-5, atol=1e-6)
