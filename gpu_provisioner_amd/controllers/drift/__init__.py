from .controller import DriftController  # noqa: F401
