"""
gordo_amd package setup.

Builds the in-tree HIP extension for MI355X (gfx950) when hipcc is
available: ``python setup.py build_ext --inplace`` produces
``gordo_amd/ops/_gordo_hip*.so`` (cross-compiles fine on GPU-less
hosts).
"""
import os
import sys

from setuptools import find_packages, setup


def _hip_extension():
    try:
        from torch.utils.cpp_extension import BuildExtension, CUDAExtension
    except ImportError:
        return [], {}
    import shutil

    if shutil.which("hipcc") is None:
        return [], {}
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    src_dir = os.path.join("gordo_amd", "ops", "csrc")
    sources = [
        os.path.join(src_dir, f)
        for f in sorted(os.listdir(src_dir))
        if f.endswith((".hip", ".cpp", ".cu"))
    ]
    if not sources:
        return [], {}
    ext = CUDAExtension(
        name="gordo_amd.ops._gordo_hip",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3", "-std=c++17"],
            "nvcc": ["-O3", "-std=c++17"],
        },
    )
    return [ext], {"build_ext": BuildExtension}


def _fastjson_extension():
    """Plain C++ (no HIP/torch) response-frame JSON encoder."""
    try:
        import pybind11
        from setuptools import Extension
    except ImportError:
        return []
    return [
        Extension(
            name="gordo_amd.server._gordo_fastjson",
            sources=[os.path.join("gordo_amd", "server", "csrc",
                                  "fastjson.cpp")],
            include_dirs=[pybind11.get_include()],
            extra_compile_args=["-O3", "-std=c++17"],
            language="c++",
        )
    ]


ext_modules, cmdclass = _hip_extension()
ext_modules += _fastjson_extension()

setup(
    name="gordo-amd",
    version="1.0.0",
    description=(
        "MI355X-native many-model timeseries anomaly engine "
        "(gordo-compatible API)"
    ),
    packages=find_packages(include=["gordo_amd", "gordo_amd.*"]),
    package_data={
        "gordo_amd.workflow.workflow_generator": ["resources/*.template"],
        "gordo_amd.ops": ["csrc/*"],
        "gordo_amd.server": ["csrc/*"],
    },
    python_requires=">=3.10",
    entry_points={
        "console_scripts": [
            "gordo=gordo_amd.cli:gordo",
            # the Argo client pods invoke `gordo-client ... predict`
            # (reference template :1375; gordo-client ships it as its
            # own package) — alias onto the in-repo client CLI
            "gordo-client=gordo_amd.cli.client:client_cli",
        ]
    },
    ext_modules=ext_modules,
    cmdclass=cmdclass,
)
