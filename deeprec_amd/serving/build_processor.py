"""Build the serving C ABI shared library (libdeeprec_processor.so).

Compiled with the system g++ against CPython headers; lands in
deeprec_amd/_ext/ so it travels to GPU boxes with the snapshot. SHA-gated
like the HIP extension build.
"""
from __future__ import annotations

import hashlib
import os
import subprocess
import sysconfig

_HERE = os.path.dirname(os.path.abspath(__file__))
_EXT_DIR = os.path.normpath(os.path.join(_HERE, "..", "_ext"))
_SRC = os.path.join(_HERE, "processor.cpp")
_SO = os.path.join(_EXT_DIR, "libdeeprec_processor.so")


def _digest() -> str:
    with open(_SRC, "rb") as f:
        return hashlib.sha256(f.read()).hexdigest()[:16]


def build_processor(force: bool = False) -> str:
    os.makedirs(_EXT_DIR, exist_ok=True)
    stamp = os.path.join(_EXT_DIR, "processor.sha")
    d = _digest()
    if (not force and os.path.exists(_SO) and os.path.exists(stamp)
            and open(stamp).read().strip() == d):
        return _SO
    inc = sysconfig.get_paths()["include"]
    libdir = sysconfig.get_config_var("LIBDIR") or "/usr/lib"
    ver = sysconfig.get_config_var("LDVERSION") or "3.10"
    cmd = ["g++", "-shared", "-fPIC", "-O2", _SRC, "-o", _SO,
           f"-I{inc}", f"-L{libdir}", f"-lpython{ver}"]
    subprocess.check_call(cmd)
    with open(stamp, "w") as f:
        f.write(d)
    return _SO


if __name__ == "__main__":
    print(build_processor())
