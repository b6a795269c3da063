from .workflow_generator import (
    get_dict_from_yaml,
    load_workflow_template,
    default_image_pull_policy,
)
from .helpers import patch_dict

__all__ = [
    "get_dict_from_yaml",
    "load_workflow_template",
    "default_image_pull_policy",
    "patch_dict",
]
