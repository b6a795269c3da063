from .from_definition import (
    from_definition,
    load_params_from_definition,
    build_callbacks,
)
from .into_definition import into_definition
from .serializer import (
    dump,
    load,
    dumps,
    loads,
    load_metadata,
    load_info,
    metadata_path,
)

__all__ = [
    "from_definition",
    "load_params_from_definition",
    "build_callbacks",
    "into_definition",
    "dump",
    "load",
    "dumps",
    "loads",
    "load_metadata",
    "load_info",
    "metadata_path",
]
