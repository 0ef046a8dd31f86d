from .errors import ApiError, ConflictError, AlreadyExistsError, NotFoundError, AdmissionDenied  # noqa: F401
from .store import InMemoryStore, WatchEvent  # noqa: F401
from .client import Client  # noqa: F401
from .workqueue import RateLimitedQueue  # noqa: F401
from .controller import Controller, Reconciler, Result, Request  # noqa: F401
from .manager import Manager  # noqa: F401
