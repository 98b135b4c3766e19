"""Training core: model/optim init, device placement, checkpoint I/O,
epoch and evaluation loops.

API-parity: /root/reference/coinstac_dinunet/nn/basetrainer.py:20-326
(NNTrainer) — same method signatures and cache-key protocol so reference
user computations drop in. MI355X redesign:
  - device placement is one-process-per-GPU (LOCAL_RANK): the reference's
    single-process torch.nn.DataParallel (basetrainer.py:62-74) is gone;
    multi-GPU scaling is the RCCL engine's job (parallel/engine.py).
  - the default optimizer is the fused multi-tensor Adam HIP kernel
    (ops.FusedAdam: one kernel launch per step over all params) on GPU,
    torch.optim.Adam on CPU — identical update math.
  - checkpoint format {'source','models':{...},'optimizers':{...}} kept,
    WITHOUT the reference's last-entry-only overwrite bug
    (basetrainer.py:104-114): every model/optimizer entry survives.
"""
import os as _os
from collections import OrderedDict as _OrderedDict

import torch as _torch

from .. import config as _conf
from .. import utils as _utils
from ..config.keys import Key
from ..metrics import COINNAverages, Prf1a
from ..utils.logger import info, success
from ..utils.tensorutils import initialize_weights as _init_weights
from ..utils.utils import performance_improved_, stop_training_

_sep = _os.sep


class NNTrainer:
    def __init__(self, data_handle=None, **kw):
        self.data_handle = data_handle
        self.cache = data_handle.cache if data_handle is not None else kw.get('cache', {})
        self.input = data_handle.input if data_handle is not None else kw.get('input', {})
        self.state = data_handle.state if data_handle is not None else kw.get('state', {})
        self.nn = _OrderedDict()
        self.device = _OrderedDict()
        self.optimizer = _OrderedDict()
        self.args = kw

    # ---- init ----------------------------------------------------------
    def _init_nn_model(self):
        raise NotImplementedError('Must be implemented by the user trainer.')

    def _init_nn_weights(self, **kw):
        if self.cache.get('pretrained_path') is not None:
            self.load_checkpoint(self.cache['pretrained_path'],
                                 load_optimizer_state=False)
        elif kw.get('init_weights'):
            _torch.manual_seed(self.cache.get('seed', _conf.current_seed))
            for mk in self.nn:
                _init_weights(self.nn[mk])

    def _init_optimizer(self):
        first_model = list(self.nn.keys())[0]
        lr = self.cache.get('learning_rate', 1e-3)
        params = self.nn[first_model].parameters()
        dev = self.device.get('gpu', _torch.device('cpu'))
        if dev.type == 'cuda':
            from .. import ops
            if ops.native_available():
                self.optimizer['adam'] = ops.FusedAdam(params, lr=lr)
                return
        self.optimizer['adam'] = _torch.optim.Adam(params, lr=lr)

    def init_nn(self, init_model=True, init_optim=True, set_devices=True,
                init_weights=False):
        if init_model:
            self._init_nn_model()
        if init_weights:
            self._init_nn_weights(init_weights=init_weights)
        if set_devices:
            self._set_gpus()
        if init_optim:
            self._init_optimizer()
        return self

    def _set_gpus(self):
        """Rank-local device placement: one process <-> one MI355X GPU."""
        gpus = self.cache.get('gpus')
        if _torch.cuda.is_available():
            if gpus:
                dev = _torch.device(f'cuda:{gpus[0]}')
            else:
                local_rank = int(_os.environ.get('LOCAL_RANK', 0))
                dev = _torch.device(f'cuda:{local_rank % _torch.cuda.device_count()}')
        else:
            dev = _torch.device('cpu')
        self.device['gpu'] = dev
        for mk in self.nn:
            self.nn[mk] = self.nn[mk].to(dev)

    # ---- checkpoint ------------------------------------------------------
    def save_checkpoint(self, file_path):
        checkpoint = {'source': 'coinstac', 'models': {}, 'optimizers': {}}
        for k, model in self.nn.items():
            model = model.module if hasattr(model, 'module') else model
            checkpoint['models'][k] = model.state_dict()
        for k, optim in self.optimizer.items():
            checkpoint['optimizers'][k] = optim.state_dict()
        _torch.save(checkpoint, file_path)

    def load_checkpoint(self, file_path, load_model_state=True,
                        load_optimizer_state=True, map_location=None):
        map_location = map_location if map_location is not None \
            else self.device.get('gpu', 'cpu')
        try:
            chk = _torch.load(file_path, map_location=map_location,
                              weights_only=False)
        except TypeError:
            chk = _torch.load(file_path, map_location=map_location)
        if isinstance(chk, dict) and chk.get('source') == 'coinstac':
            if load_model_state:
                for k, sd in chk.get('models', {}).items():
                    if k in self.nn:
                        model = self.nn[k]
                        model = model.module if hasattr(model, 'module') else model
                        model.load_state_dict(sd)
            if load_optimizer_state:
                for k, sd in chk.get('optimizers', {}).items():
                    if k in self.optimizer:
                        self.optimizer[k].load_state_dict(sd)
        else:
            # foreign plain state_dict fallback
            first = list(self.nn.keys())[0]
            model = self.nn[first]
            model = model.module if hasattr(model, 'module') else model
            try:
                model.load_state_dict(chk)
            except Exception:
                model.load_state_dict({k.replace('module.', ''): v
                                       for k, v in chk.items()})

    # ---- metric factories ------------------------------------------------
    def new_metrics(self):
        return Prf1a()

    def new_averages(self):
        return COINNAverages(num_averages=1)

    # ---- evaluation ------------------------------------------------------
    def evaluation(self, mode='eval', dataset_list=None, save_pred=False,
                   use_padded_sampler=False, **kw):
        for k in self.nn:
            self.nn[k].eval()
        # opt-in inference BN folding: evaluate against conv+BN-fused
        # copies (every BN kernel = one full activation read+write saved)
        restore = None
        if self.cache.get('fuse_bn_eval'):
            from ..ops.fuse import fuse_conv_bn_eval
            restore = dict(self.nn)
            for k in self.nn:
                self.nn[k] = fuse_conv_bn_eval(self.nn[k])
        try:
            return self._evaluation_impl(mode, dataset_list, save_pred,
                                         use_padded_sampler, **kw)
        finally:
            if restore is not None:
                self.nn.update(restore)

    def _evaluation_impl(self, mode='eval', dataset_list=None,
                         save_pred=False, use_padded_sampler=False, **kw):

        eval_avg, eval_metrics = self.new_averages(), self.new_metrics()
        eval_loaders = []
        for d in (dataset_list or []):
            if d and len(d) > 0:
                eval_loaders.append(self.data_handle.get_loader(
                    handle_key=mode, dataset=d, shuffle=False,
                    use_padded_sampler=use_padded_sampler, **kw))

        def _update(_out, _it, _avg, _metrics):
            _out = _out or {}
            if _out.get('averages', _it.get('averages')) is not None:
                _avg.accumulate(_out.get('averages', _it['averages']))
            if _out.get('metrics', _it.get('metrics')) is not None:
                _metrics.accumulate(_out.get('metrics', _it['metrics']))

        with _torch.no_grad():
            for loader in eval_loaders:
                its, avg, metrics = [], self.new_averages(), self.new_metrics()
                for i, batch in enumerate(loader, 1):
                    it = self.iteration(batch)
                    if save_pred:
                        if self.cache.get('load_sparse'):
                            its.append(it)
                        else:
                            _update(self.save_predictions(loader.dataset, it),
                                    it, avg, metrics)
                    else:
                        _update(None, it, avg, metrics)
                if save_pred and self.cache.get('load_sparse') and its:
                    its = self.reduce_iteration(its)
                    _update(self.save_predictions(loader.dataset, its),
                            its, avg, metrics)
                eval_avg.accumulate(avg)
                eval_metrics.accumulate(metrics)
        info(f'{mode} metrics: {eval_avg.get()}, {eval_metrics.get()}',
             self.cache.get('verbose'))
        return eval_avg, eval_metrics

    def save_predictions(self, dataset, its):
        """User hook: persist predictions during test; may return
        {'averages':..., 'metrics':...} to override accumulation."""
        return None

    # ---- training --------------------------------------------------------
    def training_iteration_local(self, i, batch):
        """One micro-batch: fwd/bwd; optimizer step every local_iterations."""
        it = self.iteration(batch)
        it['loss'].backward()
        if i % self.cache.get('local_iterations', 1) == 0:
            first_optim = list(self.optimizer.keys())[0]
            self.optimizer[first_optim].step()
            self.optimizer[first_optim].zero_grad()
        return it

    def reduce_iteration(self, its):
        """Merge micro-batch iteration dicts: accumulate averages/metrics,
        average scalar tensors, concat leaf tensors."""
        if len(its) == 1:
            return its[0]
        reduced = {}.fromkeys(its[0].keys(), None)
        for k in reduced:
            first = its[0][k]
            if hasattr(first, 'accumulate'):
                c = first
                for it in its[1:]:
                    c.accumulate(it[k])
                reduced[k] = c
            elif isinstance(first, _torch.Tensor) and first.dim() == 0:
                reduced[k] = sum(it[k].detach() for it in its) / len(its)
            elif isinstance(first, _torch.Tensor):
                reduced[k] = _torch.cat([it[k] for it in its])
            else:
                reduced[k] = [it[k] for it in its]
        return reduced

    def init_training_cache(self):
        self.cache[Key.TRAIN_LOG] = []
        self.cache[Key.VALIDATION_LOG] = []
        self.cache['best_val_epoch'] = 0
        self.cache['best_val_score'] = 0.0 \
            if self.cache.get('metric_direction', 'maximize') == 'maximize' \
            else _conf.max_size

    def train_local(self, train_dataset, val_dataset):
        """Full local (pre-)training loop: epoch loop with validation
        cadence, save-best and early stop."""
        out = {}
        if val_dataset is not None and not isinstance(val_dataset, list):
            val_dataset = [val_dataset]

        loader = self.data_handle.get_loader('train', dataset=train_dataset,
                                             drop_last=True, shuffle=True)
        local_iter = self.cache.get('local_iterations', 1)
        tot_iter = max(1, len(loader) // local_iter)
        ep = 0
        for ep in range(1, self.cache.get('epochs', 1) + 1):
            for k in self.nn:
                self.nn[k].train()
            _metrics, _avg = self.new_metrics(), self.new_averages()
            ep_avg, ep_metrics, its = self.new_averages(), self.new_metrics(), []
            for i, batch in enumerate(loader, 1):
                its.append(self.training_iteration_local(i, batch))
                if i % local_iter == 0:
                    it = self.reduce_iteration(its)
                    ep_avg.accumulate(it['averages'])
                    ep_metrics.accumulate(it['metrics'])
                    _avg.accumulate(it['averages'])
                    _metrics.accumulate(it['metrics'])
                    _i, its = i // local_iter, []
                    if _utils.lazy_debug(_i) or _i == tot_iter:
                        info(f"Ep:{ep}/{self.cache.get('epochs')},"
                             f"Itr:{_i}/{tot_iter},{_avg.get()},{_metrics.get()}",
                             self.cache.get('verbose'))
                        self.cache[Key.TRAIN_LOG].append(
                            [*_avg.get(), *_metrics.get()])
                        _metrics.reset()
                        _avg.reset()
                    self.on_iteration_end(i=_i, ep=ep, it=it)

            if val_dataset and ep % self.cache.get('validation_epochs', 1) == 0:
                val_averages, val_metrics = self.evaluation(
                    mode='validation', dataset_list=val_dataset,
                    use_padded_sampler=True)
                self.cache[Key.VALIDATION_LOG].append(
                    [*val_averages.get(), *val_metrics.get()])
                out.update(**self._save_if_better(ep, val_metrics))
                self._on_epoch_end(ep=ep, ep_averages=ep_avg,
                                   ep_metrics=ep_metrics,
                                   val_averages=val_averages,
                                   val_metrics=val_metrics)
                if _utils.lazy_debug(ep):
                    self._save_progress(self.cache, epoch=ep)
                if self._stop_early(ep, val_metrics, val_averages=val_averages,
                                    epoch_averages=ep_avg,
                                    epoch_metrics=ep_metrics):
                    break
        self._save_progress(self.cache, epoch=ep)
        _utils.save_cache(self.cache, self.cache.get('log_dir', '.'))
        return out

    # ---- hooks -----------------------------------------------------------
    def iteration(self, batch):
        """User hook: one mini-batch forward; returns at least
        {'loss', 'averages', 'metrics'} (and usually 'output')."""
        raise NotImplementedError

    def on_iteration_end(self, i, ep, it):
        return {}

    def _on_epoch_end(self, **kw):
        return {}

    def _save_if_better(self, epoch, val_metrics):
        out = {}
        val_score = val_metrics.extract(self.cache.get('monitor_metric', 'f1'))
        if performance_improved_(epoch, val_score, self.cache):
            self.save_checkpoint(file_path=_os.path.join(
                self.cache.get('log_dir', '.'),
                self.cache.get('best_nn_state', 'best.pt')))
            success(f'Ep {epoch}: best model saved ({val_score})',
                    self.cache.get('verbose'))
        return out

    def _save_progress(self, cache, epoch=None):
        try:
            from ..vision import plotter as _plot
            _plot.plot_progress(cache, cache.get('log_dir', '.'),
                                plot_keys=[Key.TRAIN_LOG, Key.VALIDATION_LOG],
                                epoch=epoch)
        except Exception:
            pass

    def _stop_early(self, epoch, val_metrics, **kw):
        return stop_training_(epoch, self.cache)
