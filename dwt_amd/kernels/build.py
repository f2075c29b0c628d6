"""In-tree build of the gfx950 HIP extension.

Usage: ``python -m dwt_amd.kernels.build``.  Compiles every
``src/*.hip`` + ``src/bindings.cpp`` into ``dwt_amd/kernels/_dwt_hip.so``
(the .so travels to the GPU box with the repo snapshot; a JIT cache under
~/.cache would not).  hipcc cross-compiles on a CPU-only box.
"""
from __future__ import annotations

import glob
import os
import shutil

HERE = os.path.dirname(os.path.abspath(__file__))
SRC = os.path.join(HERE, "src")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build(verbose: bool = True) -> str | None:
    sources = sorted(s for s in glob.glob(os.path.join(SRC, "*.hip"))
                     if not s.endswith("_hip.hip")) + \
        sorted(glob.glob(os.path.join(SRC, "*.cpp")))
    if not sources:
        print("dwt_amd.kernels.build: no HIP sources yet; nothing to build")
        return None

    os.environ["PYTORCH_ROCM_ARCH"] = ARCH
    os.environ.setdefault("MAX_JOBS", str(min(os.cpu_count() or 8, 16)))
    build_dir = os.path.join(HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)

    from torch.utils.cpp_extension import load

    # DWT_AMD_DEBUG=1: -O1 -g with device asserts enabled (see
    # docs/SANITIZER.md) — pair with HIP_LAUNCH_BLOCKING=1 when hunting a
    # corrupting launch
    debug = os.environ.get("DWT_AMD_DEBUG") == "1"
    opt = ["-O1", "-g"] if debug else ["-O3", "-DNDEBUG"]
    mod = load(
        name="_dwt_hip",
        sources=sources,
        extra_cflags=opt + ["-std=c++17"],
        extra_cuda_cflags=opt + ["-std=c++17"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    # copy the built module in-tree so it snapshots to the GPU box
    built = glob.glob(os.path.join(build_dir, "_dwt_hip*.so"))
    if not built:
        raise RuntimeError(f"build produced no .so under {build_dir}")
    dst = os.path.join(HERE, os.path.basename(built[0]))
    shutil.copy2(built[0], dst)
    print(f"dwt_amd.kernels.build: built {dst}")
    return dst


def main():
    build()
    import dwt_amd  # noqa: F401  (import check)
    print("dwt_amd import ok")


if __name__ == "__main__":
    main()
