from .d4pg import DDPG  # noqa: F401
from .projection import categorical_projection  # noqa: F401
from .shared_adam import SharedAdam  # noqa: F401
