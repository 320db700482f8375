from .training import train_config
from .config_utils import update_config, get_model_config

__all__ = ["train_config", "update_config", "get_model_config"]
