"""Gluon Estimator (reference gluon/contrib/estimator/estimator.py:42-430):
a batteries-included fit() loop over DataLoaders with event handlers."""
from ... import Trainer
from ...loss import Loss
from ...metric import EvalMetric, Loss as LossMetric
from .... import autograd
from ....ndarray.ndarray import NDArray
from .event_handler import (TrainBegin, TrainEnd, EpochBegin, EpochEnd,
                            BatchBegin, BatchEnd, StoppingHandler,
                            MetricHandler, LoggingHandler)

__all__ = ['Estimator']


class Estimator:
    def __init__(self, net, loss, train_metrics=None, trainer=None,
                 context=None):
        self.net = net
        self.loss = loss
        self.train_metrics = train_metrics if isinstance(train_metrics, list) \
            else ([train_metrics] if train_metrics else [])
        self.loss_metric = LossMetric(name='loss')
        self.trainer = trainer or Trainer(
            net.collect_params(), 'sgd', {'learning_rate': 0.01})
        self.context = context

    # ------------------------------------------------------------------
    def evaluate(self, val_data, batch_axis=0):
        for m in self.train_metrics:
            m.reset()
        for batch in val_data:
            x, y = batch[0], batch[1]
            pred = self.net(x)
            for m in self.train_metrics:
                m.update(y, pred)
        return {m.name: m.get()[1] for m in self.train_metrics}

    def fit_batch(self, batch, batch_axis=0):
        x, y = batch[0], batch[1]
        with autograd.record():
            pred = self.net(x)
            loss = self.loss(pred, y)
            # StochasticBlock intermediate losses (e.g. KL penalties of
            # sampled latents) join the objective (reference
            # gluon/probability StochasticBlock + estimator ELBO usage)
            extra = getattr(self.net, 'losses', None)
            if extra:
                for term in extra:
                    t = term.handle if hasattr(term, 'handle') else term
                    loss = loss + type(loss)(t.mean()) if hasattr(
                        loss, 'handle') else loss + t.mean()
        loss.backward()
        bs = x.shape[batch_axis]
        self.trainer.step(bs)
        return x, y, pred, loss

    def fit(self, train_data, val_data=None, epochs=None, event_handlers=None,
            batches=None, batch_axis=0):
        handlers = list(event_handlers or [])
        stopper = StoppingHandler(max_epoch=epochs, max_batch=batches)
        handlers.append(stopper)
        if not any(isinstance(h, MetricHandler) for h in handlers):
            handlers.append(MetricHandler([self.loss_metric]
                                          + self.train_metrics))
        if not any(isinstance(h, LoggingHandler) for h in handlers):
            handlers.append(LoggingHandler(metrics=[self.loss_metric]
                                           + self.train_metrics))

        def fire(event, *args, **kwargs):
            for h in handlers:
                fn = getattr(h, event, None)
                if fn is not None:
                    fn(self, *args, **kwargs)

        fire('train_begin')
        epoch = 0
        while not stopper.stop_training and (epochs is None or epoch < epochs):
            fire('epoch_begin')
            for batch in train_data:
                fire('batch_begin')
                x, y, pred, loss = self.fit_batch(batch, batch_axis)
                fire('batch_end', pred=pred, label=y, loss=loss)
                if stopper.stop_training:
                    break
            fire('epoch_end')
            if hasattr(train_data, 'reset'):
                train_data.reset()
            epoch += 1
            if any(getattr(h, 'stop_training', False) for h in handlers):
                break
        fire('train_end')
        return self
