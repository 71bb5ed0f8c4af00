from .cyclic import CyclicCode, build_cyclic_code
from .repetition import colocated_member_rows, group_membership, majority_vote_index
from .schedule import SEED_, AdversarySchedule

__all__ = [
    "CyclicCode",
    "build_cyclic_code",
    "colocated_member_rows",
    "group_membership",
    "majority_vote_index",
    "AdversarySchedule",
    "SEED_",
]
