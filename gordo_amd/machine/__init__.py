from .machine import Machine
from .loader import (
    load_globals_config,
    load_machine_config,
    load_model_config,
    MachineConfigException,
)

__all__ = [
    "Machine",
    "load_globals_config",
    "load_machine_config",
    "load_model_config",
    "MachineConfigException",
]
