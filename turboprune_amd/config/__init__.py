from turboprune_amd.config.compose import Config, compose, load_yaml, save_config
from turboprune_amd.config.schema import validate_config

__all__ = ["Config", "compose", "load_yaml", "save_config", "validate_config"]
