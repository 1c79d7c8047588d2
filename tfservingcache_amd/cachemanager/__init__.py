from .cachemanager import (CacheManager, make_cpu_loader,  # noqa: F401
                           make_gpu_loader)
from .lrucache import LRUCache, Model  # noqa: F401
from .modelpool import ModelPool  # noqa: F401
from .modelprovider import (ModelNotFoundError, ModelProvider,  # noqa: F401
                            ModelProviderError)
