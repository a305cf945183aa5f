from .service import ProbeSpec, ProbesService, load_probe_config

__all__ = ["ProbeSpec", "ProbesService", "load_probe_config"]
