"""pylops.optimization.basesolver.Solver stub: only the surface the
reference's CG/CGLS subclasses touch (self.Op, callback(), the print
helpers, tstart for finalize timing — ref optimization/cls_basic.py)."""
import time


class Solver:
    def __init__(self, Op, callbacks=None):
        self.Op = Op
        self.callbacks = callbacks
        self.tstart = time.time()

    def callback(self, x, *args, **kwargs):
        if self.callbacks:
            for cb in self.callbacks:
                step = getattr(cb, "on_step_end", None)
                if step is not None:
                    step(self, x)

    def _print_solver(self, nbar=65):
        print(f"{type(self).__name__}\n" + "-" * nbar)

    def _print_finalize(self, nbar=65):
        print("-" * nbar)
