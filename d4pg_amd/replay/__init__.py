from .uniform import Replay  # noqa: F401
from .per import (  # noqa: F401
    SumSegmentTree, MinSegmentTree, ReplayBuffer, PrioritizedReplayBuffer,
)
from .schedules import LinearSchedule  # noqa: F401
