from .dist import DistContext, get_context, init_from_env  # noqa: F401
from .shard import Partition, exchange_ratings_by_owner, allgather_rows  # noqa: F401
