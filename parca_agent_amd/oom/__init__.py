from .watcher import OOMWatcher, parse_oom_kill

__all__ = ["OOMWatcher", "parse_oom_kill"]
