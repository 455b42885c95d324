"""Build the hipdf native extension in-tree for gfx950.

Invoked by __graft_entry__.build() and by `python native/hipdf/build.py`.
Compiles each kernel TU with hipcc in parallel, links a pybind11 module, and
drops hipdf<ext_suffix> at the repo root so it travels with the gpurun
snapshot (JIT caches under ~/.cache do not).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
BUILD = os.path.join(HERE, "build")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNELS = [
    "kernels/elementwise.hip",
    "kernels/selection.hip",
    "kernels/scan.hip",
    "kernels/hash.hip",
    "kernels/groupby.hip",
    "kernels/join.hip",
    "kernels/partition.hip",
    "kernels/sort.hip",
    "kernels/decode.hip",
    "kernels/csv.hip",
    "kernels/strings.hip",
    "kernels/window.hip",
    "kernels/decimal128.hip",
    "kernels/regex.hip",
    "kernels/cast_str.hip",
    "kernels/datetime.hip",
    "pool.hip",
]

CXXFLAGS = ["-O3", "-std=c++20", "-fPIC", f"--offload-arch={ARCH}",
            "-Wall", "-Wno-unused-function"]


def _pybind_includes():
    import pybind11

    return [f"-I{pybind11.get_include()}",
            f"-I{sysconfig.get_paths()['include']}"]


def _newer(src, obj):
    if not os.path.exists(obj):
        return True
    common = os.path.join(HERE, "kernels")
    deps = [src] + [os.path.join(common, h)
                    for h in os.listdir(common) if h.endswith(".h")]
    return any(os.path.getmtime(d) > os.path.getmtime(obj) for d in deps)


def _compile_one(src_rel: str) -> str:
    src = os.path.join(HERE, src_rel)
    obj = os.path.join(BUILD, os.path.basename(src_rel) + ".o")
    if not _newer(src, obj):
        return obj
    cmd = [HIPCC, "-c", "-x", "hip", src, "-o", obj] + CXXFLAGS + \
        [f"-I{os.path.join(HERE, 'kernels')}"]
    subprocess.run(cmd, check=True)
    return obj


def ext_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(REPO, f"hipdf{suffix}")


def build(verbose: bool = True) -> str:
    os.makedirs(BUILD, exist_ok=True)
    if verbose:
        print(f"[hipdf] compiling {len(KERNELS)} TUs for {ARCH}")
    with ThreadPoolExecutor(max_workers=8) as pool:
        objs = list(pool.map(_compile_one, KERNELS))
    mod_src = os.path.join(HERE, "hipdf_module.cpp")
    mod_obj = os.path.join(BUILD, "hipdf_module.o")
    if _newer(mod_src, mod_obj):
        subprocess.run([HIPCC, "-c", mod_src, "-o", mod_obj] + CXXFLAGS +
                       _pybind_includes(), check=True)
    out = ext_path()
    link_inputs = objs + [mod_obj]
    if not os.path.exists(out) or any(
            os.path.getmtime(o) > os.path.getmtime(out) for o in link_inputs):
        subprocess.run([HIPCC, "-shared", "-fPIC", "-o", out] + link_inputs +
                       [f"--offload-arch={ARCH}"], check=True)
    # C ABI artifacts: libhipdf.so (kernel objects only, no python symbols)
    # + the compiled C consumer test (a non-python host can
    # link the same surface the pybind module uses)
    libso = os.path.join(REPO, "libhipdf.so")
    if not os.path.exists(libso) or any(
            os.path.getmtime(o) > os.path.getmtime(libso) for o in objs):
        subprocess.run([HIPCC, "-shared", "-fPIC", "-o", libso] + objs +
                       [f"--offload-arch={ARCH}"], check=True)
    ctest_src = os.path.join(HERE, "tests", "c_api_test.cpp")
    ctest_bin = os.path.join(REPO, "c_api_test")
    if os.path.exists(ctest_src) and (
            not os.path.exists(ctest_bin)
            or os.path.getmtime(ctest_src) > os.path.getmtime(ctest_bin)
            or os.path.getmtime(libso) > os.path.getmtime(ctest_bin)):
        subprocess.run([HIPCC, f"-I{os.path.join(HERE, 'include')}",
                        ctest_src, f"-L{REPO}", "-lhipdf",
                        f"-Wl,-rpath,$ORIGIN", "-o", ctest_bin,
                        f"--offload-arch={ARCH}"], check=True)

    # import check in a fresh interpreter (catches missing symbols at link)
    subprocess.run([sys.executable, "-c",
                    f"import sys; sys.path.insert(0, {REPO!r}); import hipdf"],
                   check=True)
    if verbose:
        print(f"[hipdf] built {out}")
    return out


if __name__ == "__main__":
    build()
