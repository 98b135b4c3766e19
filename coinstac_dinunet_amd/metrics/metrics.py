"""Metrics with cross-site serialize/reduce semantics.

Parity: /root/reference/coinstac_dinunet/metrics/metrics.py:17-329
(COINNMetrics, COINNAverages, Prf1a, ConfusionMatrix, AUCROCMetrics).
Semantics preserved exactly:
  - COINNAverages.reduce_sites SUMS values and counts (data-weighted mean);
  - score metrics' reduce_sites take the UNWEIGHTED MEAN of per-site scores.
On-GPU the confusion counts run through the HIP kernels in
coinstac_dinunet_amd.ops (single-pass atomics) when available; the
fallback is plain torch, identical numerics.
"""
import numpy as np
import torch

from ..config import metrics_eps as EPS
from ..config import metrics_num_precision as PRECISION


class COINNMetrics:
    """Interface: add(pred, true), accumulate(other), reset, get, serialize, reduce_sites."""

    def __init__(self, device='cpu', **kw):
        self.device = device
        self.eps = EPS
        self.num_precision = PRECISION

    @property
    def time(self):
        """Wall clock, for user metrics that timestamp (metrics.py:69-70)."""
        import time as _t
        return _t.time()

    def add(self, *args, **kw):
        raise NotImplementedError

    def accumulate(self, other):
        raise NotImplementedError

    def reset(self):
        raise NotImplementedError

    def get(self):
        raise NotImplementedError

    def extract(self, name):
        value = getattr(self, name.lower())  # case-tolerant (metrics.py:73)
        if callable(value):
            value = value()
        return value

    def serialize(self):
        raise NotImplementedError

    def reduce_sites(self, serialized_scores):
        """Load cross-site serialized scores INTO this instance (the remote
        calls m = trainer.new_metrics(); m.reduce_sites(scores))."""
        raise NotImplementedError

    def new(self, **kw):
        return self.__class__(device=self.device, **kw)


class COINNAverages(COINNMetrics):
    """K running (value, count) averages; reduce_sites sums -> data-weighted."""

    def __init__(self, num_averages=1, **kw):
        super().__init__(**kw)
        self.num_averages = num_averages
        self.values = np.zeros(self.num_averages, dtype=np.float64)
        self.counts = np.zeros(self.num_averages, dtype=np.int64)

    def add(self, val=0.0, n=1, index=0):
        self.values[index] += float(val) * n
        self.counts[index] += n

    def accumulate(self, other):
        self.values += other.values
        self.counts += other.counts
        return self

    def reset(self):
        self.values[...] = 0
        self.counts[...] = 0

    def get(self):
        counts = np.where(self.counts == 0, 1, self.counts)
        return [round(v, self.num_precision) for v in (self.values / counts)]

    @property
    def average(self):
        return self.get()[0]

    def serialize(self):
        return [self.values.tolist(), self.counts.tolist()]

    def reduce_sites(self, serialized):
        # Parity: metrics.py:141-143 — SUM values and counts across sites.
        for values, counts in serialized:
            values = np.asarray(values, dtype=np.float64)
            if values.shape[0] != self.num_averages:
                self.num_averages = values.shape[0]
                self.values = np.zeros(self.num_averages, dtype=np.float64)
                self.counts = np.zeros(self.num_averages, dtype=np.int64)
            self.values += values
            self.counts += np.asarray(counts, dtype=np.int64)
        return self

    def new(self, **kw):
        kw.setdefault('num_averages', self.num_averages)
        return COINNAverages(device=self.device, **kw)


def _as_long_tensor(x):
    if isinstance(x, torch.Tensor):
        return x.reshape(-1).long()
    return torch.as_tensor(np.asarray(x)).reshape(-1).long()


class Prf1a(COINNMetrics):
    """Binary precision/recall/F1/accuracy from streaming TP/FP/TN/FN counts."""

    def __init__(self, **kw):
        super().__init__(**kw)
        self.tn, self.fp, self.fn, self.tp = 0, 0, 0, 0
        # floor values populated by reduce_sites (parity: metrics.py:217-218)
        self._precision, self._recall, self._accuracy, self._f1 = 0.0, 0.0, 0.0, 0.0

    def add(self, pred, true):
        pred = _as_long_tensor(pred)
        true = _as_long_tensor(true)
        if pred.is_cuda:
            from .. import ops
            if ops.native_available():
                tp, fp, tn, fn = ops.prf1a_counts(pred, true)
                self.tp += tp; self.fp += fp; self.tn += tn; self.fn += fn
                return
        # y*2 + p case trick (parity: metrics.py:158-170)
        y_cases = true * 2 + pred
        self.tp += int((y_cases == 3).sum())
        self.fp += int((y_cases == 1).sum())
        self.tn += int((y_cases == 0).sum())
        self.fn += int((y_cases == 2).sum())

    def accumulate(self, other):
        self.tp += other.tp
        self.fp += other.fp
        self.tn += other.tn
        self.fn += other.fn
        return self

    def reset(self):
        self.tn, self.fp, self.fn, self.tp = 0, 0, 0, 0

    @property
    def precision(self):
        p = self.tp / max(self.tp + self.fp, self.eps)
        return round(max(p, self._precision), self.num_precision)

    @property
    def recall(self):
        r = self.tp / max(self.tp + self.fn, self.eps)
        return round(max(r, self._recall), self.num_precision)

    @property
    def accuracy(self):
        a = (self.tp + self.tn) / max(self.tp + self.fp + self.tn + self.fn, self.eps)
        return round(max(a, self._accuracy), self.num_precision)

    @property
    def f1(self):
        return round(max(self.f_beta(beta=1), self._f1), self.num_precision)

    def f_beta(self, beta=1):
        b2 = beta * beta
        num = (1 + b2) * self.precision * self.recall
        den = b2 * self.precision + self.recall
        return num / max(den, self.eps)

    @property
    def overlap(self):
        # IOU
        return round(self.tp / max(self.tp + self.fp + self.fn, self.eps), self.num_precision)

    def get(self):
        return [self.accuracy, self.f1, self.precision, self.recall]

    def serialize(self):
        return [self.accuracy, self.precision, self.recall]

    def reduce_sites(self, serialized):
        # Parity: metrics.py:217-218 — unweighted mean into floor values.
        arr = np.asarray([s for s in serialized if s is not None],
                         dtype=np.float64)
        if arr.size:
            acc, prec, rec = arr.mean(axis=0)
            self._accuracy, self._precision, self._recall = acc, prec, rec
            self._f1 = (2 * prec * rec) / max(prec + rec, EPS)
        return self


class ConfusionMatrix(COINNMetrics):
    """Multi-class K x K confusion matrix; macro precision/recall/F1.

    Layout: matrix[true][pred] (standard). DOCUMENTED DEVIATION: the
    reference builds matrix[pred][true] (metrics.py:243-249) yet divides
    diag by column/row sums as if rows were true — so its precision() and
    recall() are swapped relative to the standard definitions. We keep the
    standard ones; accuracy and (macro) F1 are identical either way since
    F1 is symmetric under the precision<->recall swap
    (tests/test_reference_parity.py pins this equivalence)."""

    def __init__(self, num_classes=2, **kw):
        super().__init__(**kw)
        self.num_classes = num_classes
        self.matrix = torch.zeros(num_classes, num_classes, dtype=torch.long)
        self._accuracy = 0.0
        self._prfa = None  # floor per-class [prec..., rec...] from reduce_sites

    def add(self, pred, true):
        pred = _as_long_tensor(pred)
        true = _as_long_tensor(true)
        if pred.is_cuda:
            from .. import ops
            if ops.native_available():
                self.matrix += ops.confusion_matrix(pred, true, self.num_classes).cpu()
                return
            pred, true = pred.cpu(), true.cpu()
        # sparse one-hot scatter (parity: metrics.py:243-249)
        idx = true * self.num_classes + pred
        binc = torch.bincount(idx, minlength=self.num_classes ** 2)
        self.matrix += binc.reshape(self.num_classes, self.num_classes)

    def accumulate(self, other):
        self.matrix += other.matrix
        return self

    def reset(self):
        self.matrix = torch.zeros(self.num_classes, self.num_classes, dtype=torch.long)

    @property
    def accuracy(self):
        total = self.matrix.sum().item()
        acc = self.matrix.diag().sum().item() / max(total, self.eps)
        return round(max(acc, self._accuracy), self.num_precision)

    def precision(self, average=True):
        col = self.matrix.sum(dim=0).double()
        prec = self.matrix.diag().double() / torch.clamp(col, min=self.eps)
        if self._prfa is not None:
            prec = torch.maximum(prec, torch.as_tensor(self._prfa[0], dtype=torch.double))
        return round(prec.mean().item(), self.num_precision) if average \
            else [round(v, self.num_precision) for v in prec.tolist()]

    def recall(self, average=True):
        row = self.matrix.sum(dim=1).double()
        rec = self.matrix.diag().double() / torch.clamp(row, min=self.eps)
        if self._prfa is not None:
            rec = torch.maximum(rec, torch.as_tensor(self._prfa[1], dtype=torch.double))
        return round(rec.mean().item(), self.num_precision) if average \
            else [round(v, self.num_precision) for v in rec.tolist()]

    def f1(self, average=True):
        if average:
            # reference semantics (metrics.py:269-276): harmonic mean of the
            # AVERAGED precision/recall, not the mean of per-class F1s —
            # and invariant under the reference's p<->r swap
            p, r = self.precision(True), self.recall(True)
            return round(2 * p * r / max(p + r, self.eps),
                         self.num_precision)
        p = np.asarray(self.precision(average=False))
        r = np.asarray(self.recall(average=False))
        f = (2 * p * r) / np.clip(p + r, self.eps, None)
        return [round(v, self.num_precision) for v in f.tolist()]

    def get(self):
        return [self.accuracy, self.f1(), self.precision(), self.recall()]

    def serialize(self):
        return [self.accuracy, self.precision(average=False), self.recall(average=False)]

    def reduce_sites(self, serialized):
        # Parity: metrics.py:288-289 — mean of [acc, per-class prec, per-class rec].
        serialized = [s for s in serialized if s is not None]
        if not serialized:
            return self
        accs = np.asarray([s[0] for s in serialized], dtype=np.float64)
        precs = np.asarray([s[1] for s in serialized], dtype=np.float64)
        recs = np.asarray([s[2] for s in serialized], dtype=np.float64)
        self._accuracy = float(accs.mean())
        self._prfa = (precs.mean(axis=0), recs.mean(axis=0))
        return self

    def new(self, **kw):
        kw.setdefault('num_classes', self.num_classes)
        return ConfusionMatrix(device=self.device, **kw)


class AUCROCMetrics(COINNMetrics):
    """Binary AUC from stored probabilities + labels (sklearn when available)."""

    def __init__(self, **kw):
        super().__init__(**kw)
        self.probabilities = []
        self.labels = []
        self._auc = None  # set by reduce_sites

    def add(self, prob, true):
        prob = prob.detach().reshape(-1).float().cpu().numpy() \
            if isinstance(prob, torch.Tensor) else np.asarray(prob, dtype=np.float32).reshape(-1)
        true = _as_long_tensor(true).cpu().numpy()
        self.probabilities.extend(prob.tolist())
        self.labels.extend(true.tolist())

    def accumulate(self, other):
        self.probabilities.extend(other.probabilities)
        self.labels.extend(other.labels)
        return self

    def reset(self):
        self.probabilities, self.labels = [], []
        self._auc = None

    @property
    def auc(self):
        if self._auc is not None:
            return round(self._auc, self.num_precision)
        if not self.labels or len(set(self.labels)) < 2:
            return 0.0
        try:
            from sklearn.metrics import roc_curve, auc as _auc
            fpr, tpr, _ = roc_curve(self.labels, self.probabilities)
            return round(float(_auc(fpr, tpr)), self.num_precision)
        except ImportError:
            # rank-based (Mann-Whitney) AUC with midranks for ties —
            # identical value to sklearn's trapezoidal roc_curve/auc
            y = np.asarray(self.labels)
            p = np.asarray(self.probabilities, dtype=np.float64)
            order = np.argsort(p, kind='mergesort')
            ranks = np.empty(len(p), dtype=np.float64)
            sp = p[order]
            i = 0
            while i < len(sp):
                j = i
                while j + 1 < len(sp) and sp[j + 1] == sp[i]:
                    j += 1
                ranks[order[i:j + 1]] = 0.5 * (i + j) + 1.0
                i = j + 1
            n1 = y.sum()
            n0 = len(y) - n1
            return round(float((ranks[y == 1].sum() - n1 * (n1 + 1) / 2) / (n0 * n1)),
                         self.num_precision)

    def get(self):
        return [self.auc]

    def serialize(self):
        return [self.auc]

    def reduce_sites(self, serialized):
        vals = np.asarray([s for s in serialized if s is not None],
                          dtype=np.float64).reshape(-1)
        self._auc = float(vals.mean()) if vals.size else 0.0
        return self
