"""In-tree build/load of the HIP extension for gfx950.

hipcc is invoked directly (no source translation step — the kernels are
written in HIP for CDNA4). The built .so lands in deeprec_amd/_ext/ inside
the repo so it travels to GPU boxes with the source snapshot. On a GPU box
a missing extension is a loud failure — no silent eager fallback.
"""
from __future__ import annotations

import hashlib
import importlib.machinery
import importlib.util
import os
import subprocess
import sys
import sysconfig

_EXT_NAME = "deeprec_amd_hip"
_HERE = os.path.dirname(os.path.abspath(__file__))
_EXT_DIR = os.path.normpath(os.path.join(_HERE, "..", "_ext"))
_SOURCES = [os.path.join(_HERE, "hip", "ev_kernels.hip"),
            os.path.join(_HERE, "hip", "dense_kernels.hip"),
            os.path.join(_HERE, "hip", "gru_kernels.hip"),
            os.path.join(_HERE, "hip", "attention_kernels.hip"),
            os.path.join(_HERE, "hip", "fp8_kernels.hip")]
_HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

_cached = None


def _torch_paths():
    import torch
    root = os.path.dirname(torch.__file__)
    return (
        [os.path.join(root, "include"),
         os.path.join(root, "include", "torch", "csrc", "api", "include"),
         "/opt/rocm/include",
         sysconfig.get_paths()["include"]],
        os.path.join(root, "lib"),
    )


def _so_path() -> str:
    return os.path.join(_EXT_DIR, f"{_EXT_NAME}.so")


def _sources_digest() -> str:
    h = hashlib.sha256()
    for s in _SOURCES:
        with open(s, "rb") as f:
            h.update(f.read())
    return h.hexdigest()[:16]


def build_extension(verbose: bool = False, force: bool = False):
    """Compile for gfx950 (hipcc cross-compiles without a GPU present)."""
    global _cached
    os.makedirs(_EXT_DIR, exist_ok=True)
    so = _so_path()
    stamp = os.path.join(_EXT_DIR, "source.sha")
    digest = _sources_digest()
    if (not force and os.path.exists(so) and os.path.exists(stamp)
            and open(stamp).read().strip() == digest):
        return load_extension()

    includes, libdir = _torch_paths()
    objs = []
    common_defs = [
        f"-DTORCH_EXTENSION_NAME={_EXT_NAME}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
    ]
    for src in _SOURCES:
        obj = os.path.join(
            _EXT_DIR, os.path.basename(src).replace(".hip", ".o"))
        cmd = ([_HIPCC, "-c", src, "-o", obj, "-O3", "-std=c++17", "-fPIC",
                f"--offload-arch={_ARCH}", "-fno-gpu-rdc"]
               + common_defs + [f"-isystem{p}" for p in includes])
        if verbose:
            print(" ".join(cmd))
        subprocess.run(cmd, check=True)
        objs.append(obj)
    link = ([_HIPCC, "-shared", "-o", so] + objs +
            [f"-L{libdir}", "-lc10", "-lc10_hip", "-ltorch_cpu",
             "-ltorch_hip", "-ltorch", "-ltorch_python",
             "-L/opt/rocm/lib", "-lamdhip64",
             f"-Wl,-rpath,{libdir}"])
    if verbose:
        print(" ".join(link))
    subprocess.run(link, check=True)
    with open(stamp, "w") as f:
        f.write(digest)
    _cached = None
    return load_extension()


def load_extension():
    """Load the prebuilt .so; build if absent (hipcc available everywhere)."""
    global _cached
    if _cached is not None:
        return _cached
    import torch  # noqa: F401  (extension needs torch symbols loaded)
    so = _so_path()
    if not os.path.exists(so):
        return build_extension()
    loader = importlib.machinery.ExtensionFileLoader(_EXT_NAME, so)
    spec = importlib.util.spec_from_loader(_EXT_NAME, loader)
    mod = importlib.util.module_from_spec(spec)
    loader.exec_module(mod)
    sys.modules[_EXT_NAME] = mod
    _cached = mod
    return mod


def require_extension():
    """GPU ops call this: loud failure if the native extension is missing."""
    try:
        return load_extension()
    except Exception as e:  # noqa: BLE001
        raise RuntimeError(
            "deeprec_amd HIP extension is not built/loadable. Run "
            "`python __graft_entry__.py` (build) first. Native kernels are "
            f"required on GPU — no eager fallback. Cause: {e}") from e
