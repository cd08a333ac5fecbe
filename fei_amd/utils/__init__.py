from fei_amd.utils.config import Config, get_config
from fei_amd.utils.logging import get_logger, setup_logging

__all__ = ["Config", "get_config", "get_logger", "setup_logging"]
