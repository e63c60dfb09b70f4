"""Thin inference wrapper (parity: /root/reference/npf/utils/predict.py:8-24)."""

__all__ = ["SamplePredictor"]


class SamplePredictor:
    """Multi-sample prediction with a trained NPF model; returns the predictive
    location by default, or the full distribution with `is_dist=True`."""

    def __init__(self, model, is_dist=False):
        self.model = model
        self.is_dist = is_dist

    def __call__(self, *args):
        p_y_pred, *_ = self.model(*args)
        if self.is_dist:
            return p_y_pred
        return p_y_pred.base_dist.loc.detach()
