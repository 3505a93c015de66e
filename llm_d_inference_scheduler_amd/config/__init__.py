from .loader import LoadedConfig, load_config, load_config_dict  # noqa: F401
