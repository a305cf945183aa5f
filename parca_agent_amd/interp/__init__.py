from .python import PythonUnwinder

__all__ = ["PythonUnwinder"]
