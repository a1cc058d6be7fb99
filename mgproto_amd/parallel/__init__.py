from .comm import Comm, env_rank, env_world_size, env_local_rank  # noqa: F401
from .reducer import BucketedGradReducer  # noqa: F401
from .state_sync import (DistributedEnqueue, make_dp_correct,  # noqa: F401
                         gather_push_candidates)
