from sentio_amd.engines.tokenizer import ByteTokenizer  # noqa: F401
from sentio_amd.engines.configs import MODEL_CONFIGS, ModelConfig  # noqa: F401
