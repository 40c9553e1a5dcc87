"""In-tree build of the native extensions (no JIT cache, no network).

Builds three extensions next to the package so the .so files travel with a
repo snapshot:

  gpushare_amd._amdsmi  — C++  dlopen shim over libamd_smi.so   (g++)
  gpushare_amd._devlist — C++  ListAndWatchResponse pre-encoder (g++)
  gpushare_amd._canary  — HIP  gfx950 health-probe kernels      (hipcc)
  libgpushare_memguard.so — C++ LD_PRELOAD VRAM budget enforcer (g++;
                            plain shared lib, not a Python extension)

Run:  python -m gpushare_amd.native.build  [--force]
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

import pybind11

HERE = Path(__file__).resolve().parent
PKG = HERE.parent
EXT_SUFFIX = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
GPU_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]

_COMMON = [
    "-O2",
    "-std=c++17",
    "-fPIC",
    "-shared",
    "-fvisibility=hidden",
    f"-I{pybind11.get_include()}",
    f"-I{sysconfig.get_paths()['include']}",
]


def _needs_build(src: Path, out: Path, force: bool) -> bool:
    if force or not out.exists():
        return True
    return src.stat().st_mtime > out.stat().st_mtime


def _run(cmd: list[str]) -> None:
    print("[native/build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build(force: bool = False, verbose: bool = True) -> list[Path]:
    built = []

    targets = [
        (
            "g++",
            HERE / "amdsmi_shim.cpp",
            PKG / f"_amdsmi{EXT_SUFFIX}",
            [f"-I{ROCM}/include", "-ldl"],
        ),
        ("g++", HERE / "devlist_codec.cpp", PKG / f"_devlist{EXT_SUFFIX}", []),
        (
            os.path.join(ROCM, "bin", "hipcc"),
            HERE / "canary.hip",
            PKG / f"_canary{EXT_SUFFIX}",
            [f"--offload-arch={GPU_ARCH}", "-O3"],
        ),
        (
            "g++",
            HERE / "memguard.cpp",
            PKG / "libgpushare_memguard.so",
            # visibility=default overrides _COMMON's hidden: the interposed
            # hip* symbols must be exported for LD_PRELOAD to work
            ["-fvisibility=default", "-ldl"],
        ),
    ]
    for compiler, src, out, extra in targets:
        if not _needs_build(src, out, force):
            continue
        _run([compiler, *_COMMON, str(src), "-o", str(out), *extra])
        built.append(out)
    return built


def main() -> None:
    force = "--force" in sys.argv
    built = build(force=force)
    print(f"[native/build] built {len(built)} extension(s)")
    # import check for the CPU-safe ones
    sys.path.insert(0, str(PKG.parent))
    import gpushare_amd._amdsmi as smi  # noqa: F401
    import gpushare_amd._devlist as dl  # noqa: F401

    print("[native/build] _amdsmi available:", smi.available())
    c = dl.DeviceListCodec(["amd-0-_-0"])
    assert len(c.encode()) > 0
    print("[native/build] _devlist OK")


if __name__ == "__main__":
    main()
