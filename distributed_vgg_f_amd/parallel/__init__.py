from .dist import DistContext, GradReducer, init_distributed  # noqa: F401
