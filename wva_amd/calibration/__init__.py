from .model import LlamaConfig, LlamaDecodeModel, LLAMA_3_8B, LLAMA_3_70B, TINY  # noqa: F401
from .itl_benchmark import calibrate_service_profile, fit_itl_curve  # noqa: F401
