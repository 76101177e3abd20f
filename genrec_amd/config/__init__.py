from genrec_amd.config import ginlite
from genrec_amd.config.ginlite import parse_config, configurable, bind, clear_config

__all__ = ["ginlite", "parse_config", "configurable", "bind", "clear_config"]
