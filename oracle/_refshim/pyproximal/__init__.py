"""pyproximal stub: the ProxOperator base the reference's proximal
subpackage subclasses (ref proximal/ProxOperator.py:4)."""


class ProxOperator:
    def __init__(self, Op=None, hasgrad=False):
        self.Op = Op
        self.hasgrad = hasgrad


class L1(ProxOperator):
    """Working pyproximal.L1 restatement (published closed forms:
    f(x) = sigma*||x||_1; prox_{tau f}(x) = soft(x, tau*sigma)) — the
    separable prox the reference's MPIProxOperator wraps per rank
    (ref proximal/ProxOperator.py:10-14,113-121)."""

    def __init__(self, sigma=1.0):
        super().__init__(None, False)
        self.sigma = sigma

    def __call__(self, x):
        import numpy as np
        return float(self.sigma * np.sum(np.abs(x)))

    def prox(self, x, tau):
        import numpy as np
        t = tau * self.sigma
        return np.sign(x) * np.maximum(np.abs(x) - t, 0.0)
