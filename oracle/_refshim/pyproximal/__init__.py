"""pyproximal stub: the ProxOperator base the reference's proximal
subpackage subclasses (ref proximal/ProxOperator.py:4)."""


class ProxOperator:
    def __init__(self, Op=None, hasgrad=False):
        self.Op = Op
        self.hasgrad = hasgrad
