"""torchrun entry for `gordo fleet build --gpus N` (one rank per GPU)."""
import logging
import os

from .fleet import _run_fleet_build

if __name__ == "__main__":
    logging.basicConfig(level=os.environ.get("GORDO_LOG_LEVEL", "INFO"))
    _run_fleet_build(
        os.environ["GORDO_FLEET_MACHINE_CONFIG"],
        os.environ["GORDO_FLEET_PROJECT_NAME"],
        os.environ["GORDO_FLEET_OUTPUT_DIR"],
        os.environ.get("GORDO_FLEET_MODEL_REGISTER_DIR"),
        os.environ.get("GORDO_FLEET_REPLACE_CACHE") == "1",
        os.environ.get("GORDO_FLEET_STATUS_FILE"),
    )
