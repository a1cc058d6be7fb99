"""Global configuration surface.

Mirrors the reference's two-tier config system (reference ``settings.py:1-52``):
module-level constants importable as ``from mgproto_amd import settings``, plus a
``Settings`` dataclass used by the drivers/engine so configs are overridable and
serializable (the reference hardcodes module constants only).
"""

from dataclasses import dataclass, field, asdict
from typing import Dict, List, Optional, Tuple

# ---------------------------------------------------------------------------
# Module-level constants: same names and defaults as reference settings.py
# ---------------------------------------------------------------------------

img_size = 224
num_classes = 200
prototype_shape = (num_classes * 10, 64, 1, 1)
prototype_activation_function = 'log'
add_on_layers_type = 'regular'

# Dataset roots are user-supplied; defaults point at a conventional layout.
data_path = './data/CUB_200_2011_full/'
train_dir = data_path + 'train/'
test_dir = data_path + 'test/'
train_push_dir = data_path + 'train'

data_path_ood1 = './data/Cars_full/'
test_dir_ood1 = data_path_ood1 + 'traintest/'

data_path_ood2 = './data/Pets_full/'
test_dir_ood2 = data_path_ood2 + 'traintest/'

train_batch_size = 80
test_batch_size = 80
train_push_batch_size = 80

joint_optimizer_lrs = {'features': 1e-4,
                       'add_on_layers': 3e-3,
                       'prototype_vectors': 3e-3}
joint_lr_step_size = 5

warm_optimizer_lrs = {'add_on_layers': 3e-3,
                      'prototype_vectors': 3e-3}

last_layer_optimizer_lr = 1e-4

coefs = {
    'crs_ent': 1,
    'mine': 0.2,
    'aux': 0.5,
}

num_train_epochs = 120
num_warm_epochs = 0

mine_start = 40         # 10 for the R50 schedule
updateGMM_start = 35    # 10 for the R50 schedule

push_start = 100
push_epochs = [i for i in range(num_train_epochs) if i % 10 == 0]

# LR-decay epoch lists (the reference hardcodes these per-arch in main.py:248-250)
joint_lr_decay_epochs = {
    'resnet34': [30, 45, 60, 75, 90],
    'resnet50': [10, 15, 20, 25, 30],
}
default_joint_lr_decay_epochs = [30, 45, 60, 75, 90]


# ---------------------------------------------------------------------------
# Dataclass config used by the engine / drivers
# ---------------------------------------------------------------------------

@dataclass
class Settings:
    """Structured run configuration (superset of the reference's settings.py)."""

    # model
    img_size: int = img_size
    num_classes: int = num_classes
    prototype_shape: Tuple[int, int, int, int] = prototype_shape
    prototype_activation_function: str = prototype_activation_function
    add_on_layers_type: str = add_on_layers_type
    base_architecture: str = 'resnet34'
    sz_embedding: int = 32
    mem_capacity: int = 800
    mine_K: int = 20
    aux_loss: str = 'Proxy_Anchor'

    # data
    train_dir: str = train_dir
    test_dir: str = test_dir
    train_push_dir: str = train_push_dir
    test_dir_ood1: str = test_dir_ood1
    test_dir_ood2: str = test_dir_ood2
    train_batch_size: int = train_batch_size
    test_batch_size: int = test_batch_size
    train_push_batch_size: int = train_push_batch_size
    num_workers: int = 8  # the reference ran num_workers=0 (main.py:94); we don't
    # True = one-homography fused augmentation (fast, ~4x loader throughput,
    # approximates the reference's 4-pass chain); False = the faithful
    # reference chain (main.py:98-104) for training-parity comparisons
    fast_augment: bool = True

    # optimization
    joint_optimizer_lrs: Dict[str, float] = field(default_factory=lambda: dict(joint_optimizer_lrs))
    warm_optimizer_lrs: Dict[str, float] = field(default_factory=lambda: dict(warm_optimizer_lrs))
    last_layer_optimizer_lr: float = last_layer_optimizer_lr
    coefs: Dict[str, float] = field(default_factory=lambda: dict(coefs))
    joint_lr_gamma: float = 0.4
    joint_lr_decay_epochs: Optional[List[int]] = None  # None -> per-arch table

    # schedule
    num_train_epochs: int = num_train_epochs
    num_warm_epochs: int = num_warm_epochs
    mine_start: int = mine_start
    updateGMM_start: int = updateGMM_start
    push_start: int = push_start
    push_epochs: List[int] = field(default_factory=lambda: list(push_epochs))

    # MI355X execution knobs (no counterpart in the reference)
    amp_dtype: str = 'bf16'          # backbone autocast dtype; prototype math stays fp32
    channels_last: bool = True       # NHWC layout for MIOpen convs
    em_stream: bool = True           # run the EM update on a side HIP stream
    hip_graph: bool = True           # capture the whole training step in a hipGraph
    grad_bucket_mb: int = 50         # RCCL all-reduce bucket size (xGMI-tuned)
    native_ops: str = 'auto'         # 'auto' | 'force' | 'off' — HIP kernel dispatch

    def lr_decay_epochs(self) -> List[int]:
        if self.joint_lr_decay_epochs is not None:
            return self.joint_lr_decay_epochs
        return joint_lr_decay_epochs.get(self.base_architecture,
                                         default_joint_lr_decay_epochs)

    def to_dict(self) -> dict:
        return asdict(self)
