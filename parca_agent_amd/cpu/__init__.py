from .service import CPUSamplerService

__all__ = ["CPUSamplerService"]
