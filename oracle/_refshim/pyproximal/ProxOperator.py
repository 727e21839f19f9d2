"""pyproximal.ProxOperator submodule stub: _check_tau decorator
(pyproximal's published behaviour: require tau > 0)."""
from functools import wraps


def _check_tau(func):
    @wraps(func)
    def wrapper(self, x, tau, *args, **kwargs):
        if tau <= 0:
            raise ValueError("tau must be positive")
        return func(self, x, tau, *args, **kwargs)
    return wrapper
