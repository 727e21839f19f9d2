"""pylops.utils.typing stub."""
from typing import Any, Sequence, Union

import numpy as np

NDArray = np.ndarray
DTypeLike = Any
ShapeLike = Sequence[int]
InputDimsLike = Union[int, Sequence[int]]
SamplingLike = Union[float, Sequence[float]]
