from .datastore import Datastore  # noqa: F401
