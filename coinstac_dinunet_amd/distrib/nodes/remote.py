"""Aggregator phase-machine driver.

Behavior-parity: /root/reference/coinstac_dinunet/distrib/nodes/remote.py
:22-310 — adopts shared_args from the first site, drives the reversed
fold queue, epoch/mode transitions, global score reduction, the
best-model signal and the results zip. The wire contract (out-dict keys,
phase/mode strings, quorum rules) is identical; the implementation is
organized as a table of quorum-gated phase handlers instead of the
reference's inline if-chain.
"""
import datetime as _datetime
import os as _os
import shutil as _shutil
import time as _time
import traceback as _tback

from ... import config as _conf
from ... import utils as _utils
from ...config.keys import AGG_Engine, Key, Mode, Phase
from ...utils import FrozenDict as _FrozenDict
from ...utils import lazy_debug as _lazy_debug
from ...utils.utils import performance_improved_, stop_training_
from ..reducer import COINNReducer as _dSGDReducer


class EmptyDataHandle:
    """Remote-side trainer plumbing: cache/input/state without data."""

    def __init__(self, cache, input, state):
        self.cache = cache
        self.input = input
        self.state = state


def _gather(keys, data, mode='append'):
    """Collect per-site values for `keys` across site out-dicts."""
    assert mode in ('append', 'extend'), f'Invalid gather mode: {mode}'
    res = {k: [] for k in keys}
    for d in list(data):
        for k in keys:
            value = d.get(k)
            if not value:
                continue
            if mode == 'append':
                res[k].append(value)
            else:
                res[k] = res[k] + value
    return res


def check(logic, k, v, kw):
    """Quorum predicate: logic(site[k] == v for every site input)."""
    return logic(site_vars.get(k) == v for site_vars in kw.values())


class COINNRemote:
    def __init__(self, cache=None, input=None, state=None, verbose=False, **kw):
        self.out = {}
        self.cache = cache if cache is not None else {}
        self.cache.update(**kw)
        self.input = _FrozenDict(input if input is not None else {})
        self.state = _FrozenDict(state if state is not None else {})
        self.cache['verbose'] = verbose
        self._adopt_shared_args()

    def _adopt_shared_args(self):
        """Single source of truth for hyperparameters: the sites' frozen
        args, adopted wholesale on first contact."""
        if self.cache.get(Key.ARGS_CACHED):
            return
        first_site = next(iter(self.input.values()))
        self.cache.update(**first_site['shared_args'])
        self.cache[Key.ARGS_CACHED] = True

    # =====================================================================
    # compute: quorum-gated dispatch
    # =====================================================================
    def compute(self, mp_pool, trainer_cls, reducer_cls=_dSGDReducer, **kw):
        trainer = trainer_cls(data_handle=EmptyDataHandle(
            cache=self.cache, input=self.input, state=self.state))

        self.out['phase'] = self.input.get('phase', Phase.INIT_RUNS)

        if check(all, 'phase', Phase.INIT_RUNS, self.input):
            self._handle_init_runs(trainer)
        if check(all, 'phase', Phase.PRE_COMPUTATION, self.input):
            self._handle_pre_computation()

        # every round re-publishes the per-site mode map
        self.out['global_modes'] = self._mode_map()

        if check(all, 'phase', Phase.COMPUTATION, self.input):
            self._handle_computation(trainer, reducer_cls, mp_pool)
        if check(all, 'phase', Phase.NEXT_RUN_WAITING, self.input):
            self._handle_run_end(trainer)

    # ---- INIT_RUNS ------------------------------------------------------
    def _handle_init_runs(self, trainer):
        self.cache.update(seed=self.cache.setdefault('seed',
                                                     _conf.current_seed))
        self.cache[Key.GLOBAL_TEST_SERIALIZABLE] = []
        self.cache['data_size'] = {
            site: site_vars.get('data_size')
            for site, site_vars in self.input.items()}
        # reversed so .pop() walks folds in ascending order
        self.cache['folds'] = [
            {'split_ix': str(f), 'seed': self.cache['seed']}
            for f in range(self.cache['num_folds'])][::-1]
        self.out['global_runs'] = self._open_fold(trainer)
        self.out['phase'] = Phase.NEXT_RUN

    def _open_fold(self, trainer):
        """Pop the next fold, reset best-score state, elect the pretrain
        site (the one with the most training data for this fold)."""
        fold = self.cache['fold'] = self.cache['folds'].pop()
        self.cache['log_dir'] = _os.path.join(
            self.state['outputDirectory'], self.cache['task_id'],
            f"fold_{fold['split_ix']}")
        _os.makedirs(self.cache['log_dir'], exist_ok=True)
        trainer.init_nn(set_devices=True)

        maximize = self.cache['metric_direction'] == 'maximize'
        self.cache.update(epoch=0, best_val_epoch=0,
                          best_val_score=0 if maximize else _conf.max_size)
        for log_key in (Key.TRAIN_LOG, Key.VALIDATION_LOG, Key.TEST_METRICS):
            self.cache[log_key] = []

        train_sizes = {
            site: self.cache['data_size'][site][fold['split_ix']]['train']
            for site in self.input}
        elected = max(train_sizes, key=train_sizes.get)
        return {site: {**fold, 'pretrain': site == elected}
                for site in self.input}

    # ---- PRE_COMPUTATION -------------------------------------------------
    def _handle_pre_computation(self):
        """Relay the elected site's pretrained weights to every site."""
        for site, site_vars in self.input.items():
            wfile = site_vars.get('weights_file')
            if wfile is None:
                continue
            relayed = f'pretrained_{_conf.weights_file}'
            _shutil.copy(
                _os.path.join(self.state['baseDirectory'], site, wfile),
                _os.path.join(self.state['transferDirectory'], relayed))
            self.out['pretrained_weights'] = relayed
            break
        self.out['phase'] = Phase.PRE_COMPUTATION

    # ---- COMPUTATION -----------------------------------------------------
    def _handle_computation(self, trainer, reducer_cls, mp_pool):
        reducer = self._get_reducer_cls(reducer_cls)(trainer=trainer,
                                                     mp_pool=mp_pool)
        self.out['phase'] = Phase.COMPUTATION

        if check(all, 'reduce', True, self.input):
            self.out.update(**reducer.reduce())

        if check(all, 'mode', Mode.VALIDATION_WAITING, self.input):
            # epoch boundary: validate on cadence, else straight back to train
            self.cache['epoch'] += 1
            on_cadence = (self.cache['epoch'] %
                          self.cache['validation_epochs'] == 0)
            self.out['global_modes'] = self._mode_map(
                Mode.VALIDATION if on_cadence else Mode.TRAIN)

        if check(all, 'mode', Mode.TRAIN_WAITING, self.input):
            epoch_info = self._close_epoch(reducer)
            self.out['global_modes'] = self._mode_map(
                self._next_mode(**epoch_info))

    def _close_epoch(self, reducer):
        info = self._reduce_site_scores(reducer.trainer)
        self.cache[Key.TRAIN_LOG].append(
            [*info['train_averages'].get(), *info['train_metrics'].get()])
        self._signal_save_best(**info)
        if info.get('val_averages'):
            self.cache[Key.VALIDATION_LOG].append(
                [*info['val_averages'].get(), *info['val_metrics'].get()])
        if _lazy_debug(self.cache['epoch']):
            self._plot_progress()
        return info

    def _reduce_site_scores(self, trainer):
        """Cross-site reduction of the serialized train/val scores."""
        out = {}
        for prefix, key in (('train', Key.TRAIN_SERIALIZABLE),
                            ('val', Key.VALIDATION_SERIALIZABLE)):
            rows = _gather([key], self.input.values(), 'extend')[key]
            parts = _gather(['averages', 'metrics'], rows, 'append')
            averages = trainer.new_averages()
            averages.reduce_sites(parts['averages'])
            metrics = trainer.new_metrics()
            metrics.reduce_sites(parts['metrics'])
            out[f'{prefix}_averages'] = averages
            out[f'{prefix}_metrics'] = metrics
        return out

    def _signal_save_best(self, **kw):
        if kw.get('val_metrics'):
            score = kw['val_metrics'].extract(self.cache['monitor_metric'])
            self.out['save_current_as_best'] = performance_improved_(
                self.cache['epoch'], score, self.cache)

    def _next_mode(self, **kw):
        done = self.cache['epoch'] > self.cache['epochs']
        return Mode.TEST if done or self._stop_early(**kw) else Mode.TRAIN

    def _stop_early(self, **kw):
        return stop_training_(self.cache['epoch'], self.cache)

    def _mode_map(self, mode=None):
        return {site: (mode if mode else site_vars.get('mode', 'N/A'))
                for site, site_vars in self.input.items()}

    # ---- NEXT_RUN_WAITING / SUCCESS --------------------------------------
    def _handle_run_end(self, trainer):
        self._save_fold_scores(trainer)
        if self.cache['folds']:
            self.out['global_runs'] = self._open_fold(trainer)
            self.out['phase'] = Phase.NEXT_RUN
        else:
            self.out.update(**self._publish_results(trainer))
            self.out['phase'] = Phase.SUCCESS

    def _save_fold_scores(self, trainer):
        rows = _gather([Key.TEST_SERIALIZABLE], self.input.values(),
                       'extend')[Key.TEST_SERIALIZABLE]
        parts = _gather(['averages', 'metrics'], rows, 'append')
        averages = trainer.new_averages()
        averages.reduce_sites(parts['averages'])
        metrics = trainer.new_metrics()
        metrics.reduce_sites(parts['metrics'])

        self.cache[Key.TEST_METRICS].append(
            [*averages.get(), *metrics.get()])
        self.cache[Key.GLOBAL_TEST_SERIALIZABLE].append(
            {'averages': averages.serialize(),
             'metrics': metrics.serialize()})
        self._plot_progress()
        _utils.save_scores(self.cache, self.cache['log_dir'],
                           file_keys=[Key.TEST_METRICS])
        snapshot = {**self.cache,
                    Key.GLOBAL_TEST_SERIALIZABLE:
                        self.cache[Key.GLOBAL_TEST_SERIALIZABLE][-1]}
        _utils.save_cache(snapshot, self.cache['log_dir'])

    def _publish_results(self, trainer):
        """Reduce all folds' test scores, save CSV, zip everything."""
        parts = _gather(['averages', 'metrics'],
                        self.cache[Key.GLOBAL_TEST_SERIALIZABLE], 'append')
        averages = trainer.new_averages()
        averages.reduce_sites(parts['averages'])
        metrics = trainer.new_metrics()
        metrics.reduce_sites(parts['metrics'])
        self.cache[Key.GLOBAL_TEST_METRICS] = [
            [*averages.get(), *metrics.get()]]
        task_dir = _os.path.join(self.state['outputDirectory'],
                                 self.cache['task_id'])
        _utils.save_scores(self.cache, log_dir=task_dir,
                           file_keys=[Key.GLOBAL_TEST_METRICS])

        stamp = '_'.join(str(_datetime.datetime.now()).split(' '))
        zip_name = f"{self.cache['task_id']}_{self.cache['agg_engine']}_{stamp}"
        _shutil.make_archive(
            _os.path.join(self.state['transferDirectory'], zip_name),
            'zip', task_dir)
        return {'results_zip': zip_name}

    def _plot_progress(self):
        try:
            from ...vision import plotter as _plot
            _plot.plot_progress(self.cache, self.cache['log_dir'],
                                plot_keys=[Key.TRAIN_LOG, Key.VALIDATION_LOG])
        except Exception:
            pass

    # ---- engine selection --------------------------------------------------
    def _get_reducer_cls(self, reducer_cls):
        engine = self.cache.get('agg_engine')
        if engine == AGG_Engine.dSGD:
            return _dSGDReducer
        if engine == AGG_Engine.rankDAD:
            from ..rankdad import DADReducer
            return DADReducer
        if engine == AGG_Engine.powerSGD:
            from ..powersgd import PowerSGDReducer
            return PowerSGDReducer
        return reducer_cls

    def __call__(self, *args, **kwargs):
        t0 = _time.time()
        try:
            self.compute(*args, **kwargs)
            timings = self.cache.setdefault('round_timings', [])
            timings.append([self.out.get('phase'),
                            round(_time.time() - t0, 4)])
            del timings[:-1000]
            return {'output': self.out,
                    'success': check(all, 'phase', Phase.SUCCESS, self.input)}
        except Exception:
            _tback.print_exc()
            raise Exception(self.out)
