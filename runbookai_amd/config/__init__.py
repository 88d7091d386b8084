from .schema import Config, load_config, validate_config, set_config_value

__all__ = ["Config", "load_config", "validate_config", "set_config_value"]
