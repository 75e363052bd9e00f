
from .dynamic_engine import (DynamicInferenceEngine,
                             get_dynamic_inference_engine)
