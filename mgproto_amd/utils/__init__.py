from .helpers import (list_of_distances, make_one_hot, makedir, datestr,  # noqa: F401
                      find_high_activation_crop)
from .logger import create_logger, MetricsLogger  # noqa: F401
from .memory import MemoryBank  # noqa: F401
from .receptive_field import (compute_proto_layer_rf_info_v2,  # noqa: F401
                              compute_rf_prototype, compute_rf_prototypes)
