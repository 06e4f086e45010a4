from .config import ModelConfig  # noqa: F401
