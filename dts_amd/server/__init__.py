from dts_amd.server.schemas import SearchRequest
from dts_amd.server.app import create_app

__all__ = ["SearchRequest", "create_app"]
