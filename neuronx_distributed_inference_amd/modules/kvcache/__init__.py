from neuronx_distributed_inference_amd.modules.kvcache.kv_cache_manager import (  # noqa: F401
    KVCacheManager,
)
from neuronx_distributed_inference_amd.modules.kvcache.block_kv_cache_manager import (  # noqa: F401
    BlockKVCacheManager,
)
