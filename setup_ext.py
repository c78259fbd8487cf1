"""Build modelx_amd/_core (pybind11 + HIP) in-tree for gfx950.

Drives hipcc directly (no JIT cache — the built .so lives in the repo so it
travels to GPU boxes with the source snapshot). Usage:

    python setup_ext.py build_ext --inplace
"""
import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.abspath(__file__))
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

SOURCES = [
    "core/src/engine.cpp",
    "core/hip/sha256.hip",
    "core/hip/tar.hip",
    "core/hip/dedup.hip",
    "core/hip/zstd.hip",
    "core/src/zstd_cpu.cpp",
    "core/src/http.cpp",
    "core/src/json.cpp",
]


def ext_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(REPO, "modelx_amd", "_core" + suffix)


def build() -> str:
    import pybind11

    out = ext_path()
    objs = []
    builddir = os.path.join(REPO, "build", "ext")
    os.makedirs(builddir, exist_ok=True)
    py_include = sysconfig.get_paths()["include"]
    common = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"-I{os.path.join(REPO, 'core', 'include')}",
        f"-I{pybind11.get_include()}",
        f"-I{py_include}",
        "-Wno-unused-result",
    ]
    for src in SOURCES:
        obj = os.path.join(builddir, os.path.basename(src).replace("/", "_") + ".o")
        src_abs = os.path.join(REPO, src)
        if os.path.exists(obj) and os.path.getmtime(obj) > os.path.getmtime(src_abs):
            newer_hdr = False
            hdr_dir = os.path.join(REPO, "core", "include", "modelx")
            for h in os.listdir(hdr_dir):
                if os.path.getmtime(os.path.join(hdr_dir, h)) > os.path.getmtime(obj):
                    newer_hdr = True
                    break
            if not newer_hdr:
                objs.append(obj)
                continue
        cmd = [HIPCC, "-c", src_abs, "-o", obj] + common + (["-x", "hip"] if src.endswith(".hip") else [])
        print("+", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
        objs.append(obj)
    link = [HIPCC, "-shared", "-fPIC", "-o", out] + objs + ["-lcrypto", "-lssl"]
    print("+", " ".join(link), flush=True)
    subprocess.run(link, check=True)
    return out


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "build_ext":
        build()
        print(f"built {ext_path()}")
    else:
        print(__doc__)
