from .engine import InferenceConfig, InferenceEngine, KVCache  # noqa: F401
from .serving import (ContinuousBatchingEngine, RaggedKVCache,  # noqa: F401
                      Request)
