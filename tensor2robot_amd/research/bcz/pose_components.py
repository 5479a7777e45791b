"""BC-Z pose component parameterization.

Reference `research/bcz/pose_components_lib.py:23-34`: an ActionComponent
is (name, size, is_residual, loss_weight); a StateComponent is
(name, size, is_residual).
"""

from typing import Tuple

ActionComponent = Tuple[str, int, bool, float]
StateComponent = Tuple[str, int, bool]

DEFAULT_STATE_COMPONENTS = []
DEFAULT_ACTION_COMPONENTS = [
    ("xyz", 3, True, 100.0),
    ("quaternion", 4, False, 10.0),
    ("target_close", 1, False, 1.0),
]
