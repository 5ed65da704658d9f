"""TPE — Tree-structured Parzen Estimator (BOHB-style).

Parity: /root/reference/maggy/optimizer/bayes/tpe.py:75-266 — good/bad
split at gamma=0.15, KDE density models over the transformed configs,
EI = pdf_good/pdf_bad maximized over ``n_samples`` draws from the good KDE
with bandwidth widening (bw_factor=3).  scipy.stats.gaussian_kde replaces
statsmodels KDEMultivariate (not installed).
"""
import numpy as np

from maggy_amd.optimizer.bayes.base import BaseAsyncBO


class TPE(BaseAsyncBO):
    def __init__(self, num_warmup_trials=15, random_fraction=0.33,
                 gamma=0.15, n_samples=24, bw_factor=3.0,
                 min_bandwidth=1e-3, pruner=None, pruner_kwargs=None):
        super().__init__(num_warmup_trials=num_warmup_trials,
                         random_fraction=random_fraction,
                         pruner=pruner, pruner_kwargs=pruner_kwargs)
        self.gamma = gamma
        self.n_samples = n_samples
        self.bw_factor = bw_factor
        self.min_bandwidth = min_bandwidth

    def init_model(self, budget=0):
        pass  # models are (re)built from data in update_model

    def update_model(self, budget=0):
        from scipy.stats import gaussian_kde

        X, y, n_fin = self.get_XY(budget=budget, include_busy=False)
        dim = len(self.searchspace.keys())
        # need enough points for two non-degenerate KDEs
        min_points = max(dim + 2, 4)
        if n_fin < 2 * min_points:
            return
        order = np.argsort(y)  # min convention: best first
        n_good = max(min_points, int(np.ceil(self.gamma * n_fin)))
        good = X[order[:n_good]].T
        bad = X[order[n_good:]].T
        jitter = 1e-6

        def make_kde(data):
            data = data + np.random.normal(
                0.0, jitter, size=data.shape)  # avoid singular covariance
            return gaussian_kde(data)

        try:
            self.models[budget] = {
                "good": make_kde(good), "bad": make_kde(bad)}
        except np.linalg.LinAlgError:
            return

    def sampling_routine(self, budget=0):
        model = self.models.get(budget)
        if model is None:
            return None
        kde_g, kde_b = model["good"], model["bad"]
        # draw candidates from the widened good KDE
        cand = kde_g.resample(self.n_samples)
        # bandwidth widening: extra gaussian noise scaled by bw_factor
        bw = max(kde_g.factor, self.min_bandwidth)
        cand = cand + np.random.normal(
            0.0, bw * (self.bw_factor - 1.0), size=cand.shape)
        cand = np.clip(cand, 0.0, 1.0)
        pdf_g = kde_g.pdf(cand)
        pdf_b = np.maximum(kde_b.pdf(cand), 1e-32)
        best = cand[:, int(np.argmax(pdf_g / pdf_b))]
        values = self.searchspace.inverse_transform(
            best.tolist(), normalize_categorical=True)
        return self.searchspace.list_to_dict(values)
