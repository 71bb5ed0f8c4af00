from .cyclic import CyclicCode, build_cyclic_code
from .repetition import group_membership, majority_vote_index
from .schedule import SEED_, AdversarySchedule

__all__ = [
    "CyclicCode",
    "build_cyclic_code",
    "group_membership",
    "majority_vote_index",
    "AdversarySchedule",
    "SEED_",
]
