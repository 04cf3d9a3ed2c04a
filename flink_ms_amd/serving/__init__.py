from .store import ALSModelStore, SVMModelStore, ALS_STATE_NAME, SVM_STATE_NAME  # noqa: F401
from .app import create_app  # noqa: F401
from .client import QueryClientHelper  # noqa: F401
from . import loadgen  # noqa: F401
from .sharding import ShardedQueryClient, shard_of  # noqa: F401
