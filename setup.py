"""In-tree build of the runbooks_amd gfx950 HIP extension.

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

The built `.so` lands next to the Python package (runbooks_amd/ops/) so it
travels with repo snapshots; there is no JIT cache dependency.
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension  # noqa: E402

ROOT = Path(__file__).parent
CSRC = ROOT / "runbooks_amd" / "ops" / "csrc"

sources = sorted(str(p) for p in CSRC.glob("*.hip")) + [str(CSRC / "bindings.cpp")]

setup(
    name="runbooks_amd",
    version="0.1.0",
    packages=["runbooks_amd"] + [
        f"runbooks_amd.{p}" for p in
        ("api", "cli", "client", "cloud", "controller", "k8s", "models",
         "ops", "parallel", "sci", "serve", "train", "tui", "utils",
         "workloads")],
    entry_points={"console_scripts": [
        # parity: reference cmd/ — sub CLI, controllermanager, sci servers,
        # nbwatch (goreleaser targets .goreleaser.yaml:8-37)
        "sub = runbooks_amd.cli.main:main",
        "runbooks-controller-manager = runbooks_amd.controller.manager:run_manager",
        "sci-kind = runbooks_amd.sci.kind_server:main",
        "sci-gcp = runbooks_amd.sci.gcp_server:main",
        "sci-aws = runbooks_amd.sci.aws_server:main",
        "nbwatch = runbooks_amd.nbwatch:main",
        # kubectl plugin executables (reference kubectl notebook /
        # kubectl applybuild UX — kubectl discovers kubectl-* on PATH)
        "kubectl-notebook = runbooks_amd.cli.main:kubectl_notebook",
        "kubectl-applybuild = runbooks_amd.cli.main:kubectl_applybuild",
    ]},
    ext_modules=[
        cpp_extension.CUDAExtension(
            name="runbooks_amd.ops._hip",
            sources=sources,
            extra_compile_args={
                "cxx": ["-O3", "-std=c++17"],
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    ],
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(use_ninja=True)},
)
