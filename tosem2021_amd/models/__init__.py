from .classifier import (CONFIGS, HEAD_SIZES, MLTC, MLTCConfig, build_model)
