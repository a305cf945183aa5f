from .sender import AnalyticsSender

__all__ = ["AnalyticsSender"]
