"""pyproximal.optimization.primal stub: _x0z0_init (pyproximal's
published init: x from x0, z from z0 or Op applied to x)."""


def _x0z0_init(x0, z0, Op, Opname="A"):
    if x0 is None:
        raise ValueError("x0 must be provided")
    x = x0.copy()
    if z0 is not None:
        z = z0.copy()
    elif Op is not None:
        z = Op.matvec(x)
    else:
        z = x.copy()
    return x, z
