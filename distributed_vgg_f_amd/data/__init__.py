from .synthetic import SyntheticPipeline  # noqa: F401
