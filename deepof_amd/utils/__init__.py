from .flo import read_flo, write_flo, FLO_MAGIC
from .metrics import endpoint_error, angular_error
from .flow_viz import flow_to_color

__all__ = [
    "read_flo",
    "write_flo",
    "FLO_MAGIC",
    "endpoint_error",
    "angular_error",
    "flow_to_color",
]
