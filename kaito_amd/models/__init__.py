from .registry import get_model_config, list_models, register
