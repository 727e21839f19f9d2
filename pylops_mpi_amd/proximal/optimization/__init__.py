"""Proximal solvers (ref proximal/optimization/__init__.py:1-21)."""
from .primal import ProximalGradient, ADMML2  # noqa: F401

__all__ = ["ProximalGradient", "ADMML2"]
