from .dist_context import (barrier, get_rank, get_world_size,
                           is_distributed, owned_buckets, bucket_owner)
from .exchange import exchange_by_bucket
