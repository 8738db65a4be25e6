"""Example/benchmark models built on the mpi4torch_amd primitives.

The reference ships no model zoo — its "models" are the documented
data-parallel patterns (reference examples/simple_linear_regression.py,
doc/examples.rst). These modules are those patterns as reusable code.
"""

from .linreg import DistributedLinReg
from .transformer import UlyssesTransformerBlock
from .moe import ExpertParallelMoE

__all__ = ["DistributedLinReg", "UlyssesTransformerBlock", "ExpertParallelMoE"]
