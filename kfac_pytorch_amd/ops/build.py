"""In-tree hipcc build for the gfx950 K-FAC extension.

Drives hipcc directly (no hipify, no CUDA shims) and drops the built
``_kfac_hip*.so`` next to this file so it imports as
``kfac_pytorch_amd.ops._kfac_hip`` and travels to GPU boxes with the
source tree.
"""

from __future__ import annotations

import importlib.machinery
import os
import subprocess
import sys

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
EXT_NAME = "_kfac_hip"
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

# every extension in this package: name -> (source dir, extra link libs)
EXTENSIONS = {
    "_kfac_hip": ("csrc", []),
    "_kfac_rccl": ("csrc_rccl", ["-lrccl"]),
    "_kfac_solver": ("csrc_solver", ["-lrocsolver", "-lrocblas"]),
}


def ext_path(name: str = EXT_NAME) -> str:
    suffix = importlib.machinery.EXTENSION_SUFFIXES[0]
    return os.path.join(OPS_DIR, name + suffix)


def _sources(name: str = EXT_NAME):
    src_dir = os.path.join(OPS_DIR, EXTENSIONS[name][0])
    return [os.path.join(src_dir, f) for f in sorted(os.listdir(src_dir))
            if f.endswith(".hip")]


def needs_build(name: str = EXT_NAME) -> bool:
    so = ext_path(name)
    if not os.path.exists(so):
        return True
    so_mtime = os.path.getmtime(so)
    return any(os.path.getmtime(s) > so_mtime
               for s in _sources(name) + [os.path.abspath(__file__)])


def build_all(verbose: bool = True, force: bool = False):
    """Compile every extension for gfx950. Returns the .so paths."""
    return [build(verbose=verbose, force=force, name=n) for n in EXTENSIONS]


def build(verbose: bool = True, force: bool = False,
          name: str = EXT_NAME) -> str:
    """Compile one extension with hipcc for gfx950. Returns the .so path."""
    so = ext_path(name)
    if not force and not needs_build(name):
        return so

    import torch
    import torch.utils.cpp_extension as ce

    import sysconfig
    py_include = sysconfig.get_paths()["include"]

    cmd = [
        os.path.join(ce.ROCM_HOME or "/opt/rocm", "bin", "hipcc"),
        f"--offload-arch={GFX_ARCH}",
        "-O3", "-std=c++17", "-shared", "-fPIC",
        f"-DTORCH_EXTENSION_NAME={name}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={int(torch._C._GLIBCXX_USE_CXX11_ABI)}",
    ]
    cmd += ce.COMMON_HIP_FLAGS + ce.COMMON_HIPCC_FLAGS
    for inc in ce.include_paths("cuda"):
        cmd.append(f"-I{inc}")
    cmd.append(f"-I{py_include}")
    cmd += _sources(name)
    for lp in ce.library_paths("cuda"):
        cmd.append(f"-L{lp}")
    cmd += ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python",
            "-lamdhip64"] + EXTENSIONS[name][1] + ["-o", so]

    if verbose:
        print(f"[kfac build] hipcc -> {os.path.basename(so)} "
              f"({GFX_ARCH})", flush=True)
    try:
        subprocess.run(cmd, check=True, capture_output=True, text=True)
    except subprocess.CalledProcessError as e:
        sys.stderr.write(e.stdout or "")
        sys.stderr.write(e.stderr or "")
        raise RuntimeError(
            f"hipcc build of {name} failed (see output above)") from e
    return so


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
