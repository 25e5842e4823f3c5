from .fanout import (broadcast_blob, init_distributed,  # noqa: F401
                     shard_assignment, sharded_pull_fanout)
