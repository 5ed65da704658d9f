"""GP-based async Bayesian optimizer.

Parity: /root/reference/maggy/optimizer/bayes/gp.py:100-373 — Gaussian
process surrogate (ConstantKernel x Matern nu=2.5, as the reference builds
via skopt), busy-location imputation by constant liar (cl_min/cl_max/
cl_mean) or kriging believer (the model's own posterior mean, reference
gp.py:329-373), acquisition EI/LCB/PI optimized by dense random sampling
followed by L-BFGS-B refinement from the best candidates (the reference's
acq_optimizer="lbfgs" path with n_restarts, gp.py:183-264).
sklearn replaces skopt (not installed); the model is equivalent.
"""
import numpy as np

from maggy_amd.optimizer.bayes.base import BaseAsyncBO


class GP(BaseAsyncBO):
    def __init__(self, num_warmup_trials=15, random_fraction=0.33,
                 acq_fun="EI", acq_n_points=10000, acq_n_restarts=5,
                 xi=0.01, kappa=1.96,
                 async_strategy="impute", imputed_metric="cl_min",
                 interim_results=False, interim_results_interval=3,
                 pruner=None, pruner_kwargs=None):
        super().__init__(num_warmup_trials=num_warmup_trials,
                         random_fraction=random_fraction,
                         interim_results=interim_results,
                         interim_results_interval=interim_results_interval,
                         pruner=pruner, pruner_kwargs=pruner_kwargs)
        if acq_fun not in ("EI", "PI", "LCB"):
            raise ValueError("acq_fun must be EI, PI or LCB")
        if async_strategy not in ("impute", "asy_ts"):
            raise ValueError("async_strategy must be 'impute' or 'asy_ts'")
        if imputed_metric not in ("cl_min", "cl_max", "cl_mean", "kb"):
            raise ValueError(
                "imputed_metric must be cl_min, cl_max, cl_mean or kb")
        self.acq_fun = acq_fun
        self.acq_n_points = acq_n_points
        self.acq_n_restarts = acq_n_restarts
        self.xi = xi
        self.kappa = kappa
        self.async_strategy = async_strategy
        self.imputed_metric = imputed_metric
        self._y_fit_min = {}  # budget -> incumbent in ORIGINAL scale

    def _make_gp(self):
        from sklearn.gaussian_process import GaussianProcessRegressor
        from sklearn.gaussian_process.kernels import (
            ConstantKernel,
            Matern,
        )

        dim = len(self.searchspace.keys()) + (
            1 if self.interim_results else 0)
        kernel = ConstantKernel(1.0) * Matern(
            length_scale=np.ones(dim), nu=2.5)
        return GaussianProcessRegressor(
            kernel=kernel, normalize_y=True, alpha=1e-6,
            n_restarts_optimizer=2, random_state=0)

    def init_model(self, budget=0):
        self.models[budget] = self._make_gp()

    def update_model(self, budget=0):
        X, y, n_fin = self.get_XY(
            budget=budget,
            include_busy=(self.async_strategy == "impute"))
        if n_fin < 2:
            return
        if budget not in self.models:
            self.init_model(budget)
        import warnings

        with warnings.catch_warnings():
            # kernel-hyperparameter optimizer convergence chatter is normal
            # on small async batches
            warnings.simplefilter("ignore")
            self.models[budget].fit(X, y)
        # incumbent in ORIGINAL scale: with normalize_y=True sklearn stores
        # y_train_ normalized while predict() returns original-scale mu, so
        # min(y_train_) would mix scales and degenerate EI/PI;
        # only the first n_fin entries of y are real observations (the rest
        # are imputed busy locations)
        self._y_fit_min[budget] = float(np.min(y[:n_fin]))

    def sampling_routine(self, budget=0):
        model = self.models.get(budget)
        if model is None or not hasattr(model, "X_train_"):
            return None
        dim = len(self.searchspace.keys())
        cand = np.random.uniform(0.0, 1.0, size=(self.acq_n_points, dim))
        if self.interim_results:
            # evaluate candidates at full progress
            cand = np.concatenate(
                [cand, np.ones((self.acq_n_points, 1))], axis=1)
        if self.async_strategy == "asy_ts":
            # asynchronous Thompson sampling: one posterior draw, minimize
            sample = model.sample_y(cand, n_samples=1,
                                    random_state=None).ravel()
            best = cand[int(np.argmin(sample))]
        else:
            y_best = self._y_fit_min.get(budget)
            if y_best is None:
                return None
            mu, sigma = model.predict(cand, return_std=True)
            score = self._acquisition(mu, sigma, y_best)
            order = np.argsort(score)[::-1]
            best = cand[int(order[0])]
            best = self._refine_lbfgs(
                model, y_best, cand[order[: self.acq_n_restarts]], best)
        values = self.searchspace.inverse_transform(
            best[:dim].tolist(), normalize_categorical=True)
        return self.searchspace.list_to_dict(values)

    def _refine_lbfgs(self, model, y_best, starts, best):
        """Polish the top sampled candidates with L-BFGS-B over the unit
        hypercube (parity: reference gp.py:183-264 acq_optimizer='lbfgs'
        with n_restarts starts from sampled points)."""
        from scipy.optimize import minimize

        def neg_acq(x):
            mu, sigma = model.predict(x.reshape(1, -1), return_std=True)
            return -float(self._acquisition(mu, sigma, y_best)[0])

        best_val = neg_acq(best)
        bounds = [(0.0, 1.0)] * starts.shape[1]
        if self.interim_results:
            bounds[-1] = (1.0, 1.0)  # progress coordinate stays at "done"
        for x0 in starts:
            try:
                res = minimize(neg_acq, x0, method="L-BFGS-B", bounds=bounds,
                               options={"maxiter": 20})
            except Exception as e:
                self._log("L-BFGS-B refinement failed: {}".format(e))
                continue
            if res.fun < best_val:
                best_val = res.fun
                best = np.clip(res.x, 0.0, 1.0)
        return best

    def _acquisition(self, mu, sigma, y_best):
        from scipy.stats import norm

        sigma = np.maximum(sigma, 1e-9)
        if self.acq_fun == "LCB":
            return -(mu - self.kappa * sigma)
        imp = y_best - mu - self.xi
        z = imp / sigma
        if self.acq_fun == "EI":
            return imp * norm.cdf(z) + sigma * norm.pdf(z)
        return norm.cdf(z)  # PI
