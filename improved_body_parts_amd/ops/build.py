"""In-tree build of the _ibp_hip extension for gfx950.

``python -m improved_body_parts_amd.ops.build`` compiles every ``csrc/*.hip`` +
``csrc/bindings.cpp`` into ``improved_body_parts_amd/ops/_ibp_hip.so`` next to
this file, so the built artefact travels with the repo snapshot to GPU boxes.
hipcc cross-compiles for gfx950 without a GPU present.
"""
from __future__ import annotations

import glob
import os
import sys

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")
OUT_SO = os.path.join(PKG_DIR, "_ibp_hip.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _sources():
    # torch's hipify pass writes shadow copies named *_hip.hip next to the
    # originals; exclude them or a rebuild would compile every file twice.
    hips = [s for s in sorted(glob.glob(os.path.join(CSRC, "*.hip")))
            if not s.endswith("_hip.hip")]
    return hips + sorted(glob.glob(os.path.join(CSRC, "*.cpp")))


def _needs_rebuild(sources):
    if not os.path.exists(OUT_SO):
        return True
    so_mtime = os.path.getmtime(OUT_SO)
    deps = sources + glob.glob(os.path.join(CSRC, "*.h"))
    return any(os.path.getmtime(s) > so_mtime for s in deps)


def build_extension(verbose: bool = True, force: bool = False) -> str | None:
    sources = _sources()
    if not sources:
        if verbose:
            print("build_extension: no csrc sources yet; nothing to build")
        return None
    if not force and not _needs_rebuild(sources):
        if verbose:
            print(f"build_extension: {OUT_SO} is up to date")
        return OUT_SO

    os.environ.setdefault("PYTORCH_ROCM_ARCH", ARCH)
    from torch.utils import cpp_extension

    build_dir = os.path.join(CSRC, "build")
    os.makedirs(build_dir, exist_ok=True)
    module = cpp_extension.load(
        name="_ibp_hip",
        sources=sources,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", f"--offload-arch={ARCH}",
                           "-DNDEBUG"],
        build_directory=build_dir,
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    # cpp_extension.load writes the .so into build_dir; copy it in-tree
    built = os.path.join(build_dir, "_ibp_hip.so")
    if os.path.exists(built):
        import shutil
        shutil.copy2(built, OUT_SO)
        if verbose:
            print(f"build_extension: wrote {OUT_SO}")
        return OUT_SO
    raise RuntimeError(f"extension build produced no .so in {build_dir}")


if __name__ == "__main__":
    build_extension(force="--force" in sys.argv)
