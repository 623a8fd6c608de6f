"""Dynamic loss scaler (reference python/mxnet/amp/loss_scaler.py)."""


class LossScaler:
    def __init__(self, init_scale=2 ** 16, scale_factor=2.0,
                 scale_window=2000, min_scale=1.0):
        self.loss_scale = float(init_scale)
        self._scale_factor = scale_factor
        self._scale_window = scale_window
        self._min_scale = min_scale
        self._unskipped = 0

    def has_overflow(self, params):
        """Check grads for inf/nan using the fused all_finite kernel."""
        from . import all_finite
        grads = []
        for p in params:
            try:
                grads.extend(p.list_grad())
            except Exception:
                pass
        return not all_finite(grads)

    def update_scale(self, overflow):
        if overflow:
            self.loss_scale = max(self.loss_scale / self._scale_factor,
                                  self._min_scale)
            self._unskipped = 0
        else:
            self._unskipped += 1
            if self._unskipped >= self._scale_window:
                self.loss_scale *= self._scale_factor
                self._unskipped = 0
        return self.loss_scale
