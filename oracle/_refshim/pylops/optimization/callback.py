"""pylops.optimization.callback stub (imported by the reference's
sparsity module; not exercised by the ref-parity suites)."""


class Callbacks:
    def on_setup_begin(self, solver, x0):
        pass

    def on_setup_end(self, solver, x):
        pass

    def on_step_begin(self, solver, x):
        pass

    def on_step_end(self, solver, x):
        pass

    def on_run_begin(self, solver, x):
        pass

    def on_run_end(self, solver, x):
        pass


class CostNanInfCallback(Callbacks):
    pass


class CostToInitialCallback(Callbacks):
    pass


class CostToDataCallback(Callbacks):
    pass
