from byzpy_amd.attacks.base import Attack
from byzpy_amd.attacks.ops import (
    EmpireAttack,
    GaussianAttack,
    InfAttack,
    LabelFlipAttack,
    LittleAttack,
    MimicAttack,
    SignFlipAttack,
)

__all__ = [
    "Attack",
    "EmpireAttack",
    "SignFlipAttack",
    "LabelFlipAttack",
    "LittleAttack",
    "GaussianAttack",
    "InfAttack",
    "MimicAttack",
]
