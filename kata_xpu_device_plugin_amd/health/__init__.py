from .watcher import NodeWatcher  # noqa: F401
