from .state import GmmState  # noqa: F401
