"""Evaluation metrics (reference python/mxnet/gluon/metric.py)."""
import numpy as _np

from ..ndarray.ndarray import NDArray

_REGISTRY = {}


def register(cls):
    _REGISTRY[cls.__name__.lower()] = cls
    return cls


def create(name, **kwargs):
    if isinstance(name, EvalMetric):
        return name
    return _REGISTRY[name.lower()](**kwargs)


def _np_of(x):
    if isinstance(x, NDArray):
        return x.asnumpy()
    return _np.asarray(x)


class EvalMetric:
    def __init__(self, name, output_names=None, label_names=None, **kwargs):
        self.name = name
        self.reset()

    def update(self, labels, preds):
        raise NotImplementedError

    def reset(self):
        self.num_inst = 0
        self.sum_metric = 0.0

    def get(self):
        if self.num_inst == 0:
            return (self.name, float('nan'))
        return (self.name, self.sum_metric / self.num_inst)

    def get_name_value(self):
        name, value = self.get()
        if not isinstance(name, list):
            name, value = [name], [value]
        return list(zip(name, value))


@register
class Accuracy(EvalMetric):
    def __init__(self, axis=1, name='accuracy', **kwargs):
        self.axis = axis
        super().__init__(name, **kwargs)

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            pred = _np_of(pred)
            label = _np_of(label)
            if pred.ndim > label.ndim:
                pred = pred.argmax(axis=self.axis)
            self.sum_metric += (pred.astype('int64').flat ==
                                label.astype('int64').flat).sum()
            self.num_inst += label.size


@register
class TopKAccuracy(EvalMetric):
    def __init__(self, top_k=1, name='top_k_accuracy', **kwargs):
        self.top_k = top_k
        super().__init__(f'{name}_{top_k}', **kwargs)

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            pred = _np_of(pred)
            label = _np_of(label).astype('int64')
            topk = _np.argsort(-pred, axis=-1)[:, :self.top_k]
            self.sum_metric += (topk == label[:, None]).any(axis=1).sum()
            self.num_inst += label.shape[0]


@register
class MAE(EvalMetric):
    def __init__(self, name='mae', **kwargs):
        super().__init__(name, **kwargs)

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            label, pred = _np_of(label), _np_of(pred)
            self.sum_metric += _np.abs(label - pred.reshape(label.shape)).sum()
            self.num_inst += label.size


@register
class MSE(EvalMetric):
    def __init__(self, name='mse', **kwargs):
        super().__init__(name, **kwargs)

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            label, pred = _np_of(label), _np_of(pred)
            self.sum_metric += ((label - pred.reshape(label.shape)) ** 2).sum()
            self.num_inst += label.size


@register
class RMSE(MSE):
    def __init__(self, name='rmse', **kwargs):
        EvalMetric.__init__(self, name, **kwargs)

    def get(self):
        if self.num_inst == 0:
            return (self.name, float('nan'))
        return (self.name, _np.sqrt(self.sum_metric / self.num_inst))


@register
class CrossEntropy(EvalMetric):
    def __init__(self, eps=1e-12, name='cross-entropy', **kwargs):
        self.eps = eps
        super().__init__(name, **kwargs)

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            label = _np_of(label).astype('int64').ravel()
            pred = _np_of(pred)
            prob = pred[_np.arange(label.shape[0]), label]
            self.sum_metric += (-_np.log(prob + self.eps)).sum()
            self.num_inst += label.shape[0]


@register
class Perplexity(CrossEntropy):
    def __init__(self, ignore_label=None, name='perplexity', **kwargs):
        self.ignore_label = ignore_label
        EvalMetric.__init__(self, name, **kwargs)
        self.eps = 1e-12

    def get(self):
        if self.num_inst == 0:
            return (self.name, float('nan'))
        return (self.name, _np.exp(self.sum_metric / self.num_inst))


@register
class F1(EvalMetric):
    def __init__(self, name='f1', average='macro', **kwargs):
        self.average = average
        super().__init__(name, **kwargs)

    def reset(self):
        super().reset()
        self.tp = self.fp = self.fn = 0

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            pred = _np_of(pred)
            label = _np_of(label).astype('int64').ravel()
            if pred.ndim > 1:
                pred = pred.argmax(axis=-1)
            pred = pred.astype('int64').ravel()
            self.tp += int(((pred == 1) & (label == 1)).sum())
            self.fp += int(((pred == 1) & (label == 0)).sum())
            self.fn += int(((pred == 0) & (label == 1)).sum())
            self.num_inst += label.size

    def get(self):
        prec = self.tp / max(self.tp + self.fp, 1)
        rec = self.tp / max(self.tp + self.fn, 1)
        f1 = 2 * prec * rec / max(prec + rec, 1e-12)
        return (self.name, f1)


@register
class MCC(EvalMetric):
    def __init__(self, name='mcc', **kwargs):
        super().__init__(name, **kwargs)

    def reset(self):
        super().reset()
        self.tp = self.fp = self.fn = self.tn = 0

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            pred = _np_of(pred)
            label = _np_of(label).astype('int64').ravel()
            if pred.ndim > 1:
                pred = pred.argmax(axis=-1)
            pred = pred.astype('int64').ravel()
            self.tp += int(((pred == 1) & (label == 1)).sum())
            self.fp += int(((pred == 1) & (label == 0)).sum())
            self.fn += int(((pred == 0) & (label == 1)).sum())
            self.tn += int(((pred == 0) & (label == 0)).sum())
            self.num_inst += label.size

    def get(self):
        num = self.tp * self.tn - self.fp * self.fn
        den = _np.sqrt(float((self.tp + self.fp) * (self.tp + self.fn)
                             * (self.tn + self.fp) * (self.tn + self.fn)))
        return (self.name, num / den if den else 0.0)


@register
class PearsonCorrelation(EvalMetric):
    def __init__(self, name='pearsonr', **kwargs):
        super().__init__(name, **kwargs)

    def reset(self):
        super().reset()
        self._labels, self._preds = [], []

    def update(self, labels, preds):
        if isinstance(labels, (NDArray, _np.ndarray)):
            labels, preds = [labels], [preds]
        for label, pred in zip(labels, preds):
            self._labels.append(_np_of(label).ravel())
            self._preds.append(_np_of(pred).ravel())
            self.num_inst += 1

    def get(self):
        if not self._labels:
            return (self.name, float('nan'))
        l = _np.concatenate(self._labels)
        p = _np.concatenate(self._preds)
        return (self.name, float(_np.corrcoef(l, p)[0, 1]))


@register
class Loss(EvalMetric):
    def __init__(self, name='loss', **kwargs):
        super().__init__(name, **kwargs)

    def update(self, _, preds):
        if isinstance(preds, (NDArray, _np.ndarray)):
            preds = [preds]
        for pred in preds:
            loss = _np_of(pred)
            self.sum_metric += loss.sum()
            self.num_inst += loss.size


class CompositeEvalMetric(EvalMetric):
    def __init__(self, metrics=None, name='composite', **kwargs):
        super().__init__(name, **kwargs)
        self.metrics = [create(m) if isinstance(m, str) else m
                        for m in (metrics or [])]

    def add(self, metric):
        self.metrics.append(create(metric) if isinstance(metric, str) else metric)

    def update(self, labels, preds):
        for m in self.metrics:
            m.update(labels, preds)

    def reset(self):
        for m in getattr(self, 'metrics', []):
            m.reset()

    def get(self):
        names, values = [], []
        for m in self.metrics:
            n, v = m.get()
            names.append(n)
            values.append(v)
        return (names, values)
