"""Build the gfx950 HIP extension in-tree.

``python -m lpp_amd.ops.build`` (or ``__graft_entry__.build()``) compiles
every ``.hip``/``.cpp`` under ``lpp_amd/ops/csrc`` with hipcc for
``--offload-arch=gfx950`` into ``lpp_amd/ops/_lpp_kernels.so``.  The build
is in-tree so the shared object travels with the repo snapshot to GPU boxes
(no JIT cache dependence).
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

CSRC = Path(__file__).resolve().parent / "csrc"
OUT_DIR = Path(__file__).resolve().parent
SO_NAME = "_lpp_kernels.so"
MODULE_NAME = "_lpp_kernels"


def _sources():
    return sorted([*CSRC.glob("*.hip"), *CSRC.glob("*.cpp")])


def _newest_mtime(paths):
    return max((p.stat().st_mtime for p in paths), default=0.0)


def build(verbose: bool = True, force: bool = False) -> Path:
    """Compile the extension with hipcc (cross-compiles fine without a GPU)."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    sources = _sources()
    if not sources:
        raise RuntimeError(f"no HIP sources under {CSRC}")
    so_path = OUT_DIR / SO_NAME
    if so_path.exists() and not force and so_path.stat().st_mtime >= _newest_mtime(sources):
        return so_path

    import torch
    from torch.utils import cpp_extension as ce

    torch_lib = Path(torch.__file__).parent / "lib"
    rocm = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
    hipcc = str(rocm / "bin" / "hipcc")

    include_dirs = ce.include_paths() + [str(CSRC)]
    objs = []
    build_dir = OUT_DIR / "_build"
    build_dir.mkdir(exist_ok=True)
    for src in sources:
        obj = build_dir / (src.stem + ".o")
        cmd = [
            hipcc,
            "-O3",
            "-std=c++17",
            "-fPIC",
            "--offload-arch=gfx950",
            "-DUSE_ROCM",
            "-D__HIP_PLATFORM_AMD__",
            f"-DTORCH_EXTENSION_NAME={MODULE_NAME}",
            "-D_GLIBCXX_USE_CXX11_ABI=" + ("1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"),
            "-fno-gpu-rdc",
            "-c",
            str(src),
            "-o",
            str(obj),
        ]
        for inc in include_dirs:
            cmd += ["-I", inc]
        # python headers
        import sysconfig

        cmd += ["-I", sysconfig.get_paths()["include"]]
        if verbose:
            print("[lpp build]", " ".join(cmd), flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed for {src.name}:\n{r.stdout}\n{r.stderr}")
        if r.stderr and verbose:
            sys.stderr.write(r.stderr)
        objs.append(obj)

    link = [
        hipcc,
        "-shared",
        "-fPIC",
        *map(str, objs),
        "-o",
        str(so_path),
        f"-L{torch_lib}",
        "-ltorch",
        "-ltorch_cpu",
        "-ltorch_python",
        "-lc10",
        f"-L{rocm}/lib",
        "-lamdhip64",
        "-lhipblaslt",
    ]
    # torch hip libs
    if (torch_lib / "libtorch_hip.so").exists():
        link.insert(-3, "-ltorch_hip")
        link.insert(-3, "-lc10_hip")
    if verbose:
        print("[lpp build]", " ".join(link), flush=True)
    r = subprocess.run(link, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    return so_path


def load_extension():
    """Import the built extension (build first if sources are newer)."""
    so_path = OUT_DIR / SO_NAME
    if not so_path.exists():
        build(verbose=False)
    import importlib.util

    spec = importlib.util.spec_from_file_location(MODULE_NAME, so_path)
    mod = importlib.util.module_from_spec(spec)
    # The extension links against libtorch*, which the torch package has
    # already loaded into the process.
    import torch  # noqa: F401

    spec.loader.exec_module(mod)
    return mod


if __name__ == "__main__":
    p = build(verbose=True, force="--force" in sys.argv)
    print(f"built {p}")
