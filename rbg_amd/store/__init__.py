from .store import Store, Event, Watch, Conflict, NotFound, AlreadyExists, set_owner  # noqa: F401
from .revisions import RevisionManager, hash_spec  # noqa: F401
