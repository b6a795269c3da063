"""
gordo_amd — an MI355X-native many-model timeseries anomaly engine.

A from-scratch reimplementation of the capabilities of equinor/gordo
(reference: /root/reference) designed for AMD Instinct MI355X (gfx950):
thousands of small per-asset autoencoders are packed into grouped MFMA
GEMM batches on each GPU, machines are sharded across the 8 GPUs of a
node with torch.distributed over RCCL/xGMI, and the serving hot path is
fused HIP scoring kernels.

Public surface mirrors the reference (``gordo`` package): YAML Machine
configs, sklearn-Pipeline serializer, ``gordo build`` / ``gordo
run-server`` / ``gordo workflow generate`` CLIs and the
``/prediction`` + ``/anomaly/prediction`` HTTP API.
(Reference version parsing: gordo/__init__.py:15-68.)
"""
import re

__version__ = "1.0.0"


def parse_version(version: str):
    """
    Parse a version string into (major, minor) and whether it is an
    "unstable" (dev/pre-release) version.

    Examples
    --------
    >>> parse_version("1.2.3")
    (1, 2, False)
    >>> parse_version("1.2.3.dev4+gf1a2b3c")
    (1, 2, True)
    """
    parts = version.split(".")
    if len(parts) < 2:
        raise ValueError(f"Malformed version {version!r}")
    try:
        major, minor = int(parts[0]), int(parts[1])
    except ValueError as e:
        raise ValueError(f"Malformed version {version!r}") from e
    unstable = bool(re.search(r"(dev|rc|a|b)\d*", ".".join(parts[2:]))) or (
        "+" in version
    )
    return major, minor, unstable


MAJOR_VERSION, MINOR_VERSION, IS_UNSTABLE_VERSION = parse_version(__version__)
