from .objects import KfObject, Condition, new_object, set_condition, get_condition
from .store import ObjectStore, Event, ConflictError, NotFoundError, AlreadyExistsError

__all__ = [
    "KfObject", "Condition", "new_object", "set_condition", "get_condition",
    "ObjectStore", "Event", "ConflictError", "NotFoundError",
    "AlreadyExistsError",
]
