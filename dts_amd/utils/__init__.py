from dts_amd.utils.logging import logger

__all__ = ["logger"]
