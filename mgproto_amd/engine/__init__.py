from .trainer import (train, test, warm_only, joint, EMRunner,  # noqa: F401
                      _training, _testing, _testing_with_OoD)
from .push import push_prototypes  # noqa: F401
