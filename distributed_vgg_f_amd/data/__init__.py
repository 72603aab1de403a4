from .synthetic import SyntheticPipeline  # noqa: F401
from .records import RecordPipeline, write_shards  # noqa: F401
