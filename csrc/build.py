#!/usr/bin/env python3
"""Build the in-tree HIP extension: csrc/*.hip + bindings.cpp -> csrc/_dmnist_hip.so

Direct hipcc invocation (gfx950 only, no GPU needed to compile). The .so is
committed-adjacent (git-ignored) and travels with gpurun snapshots.
"""

import os
import subprocess
import sys
import sysconfig

ROOT = os.path.dirname(os.path.abspath(__file__))


def main():
    import torch
    import torch.utils.cpp_extension as ce

    torch_inc = ce.include_paths()
    torch_lib = ce.library_paths()[0]
    py_inc = sysconfig.get_paths()["include"]
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    abi = int(torch.compiled_with_cxx11_abi())

    srcs = [os.path.join(ROOT, f) for f in
            ("gemm_tile.hip", "conv_slab.hip", "dw_tr.hip", "ops_misc.hip", "bindings.cpp")]
    out = os.path.join(ROOT, "_dmnist_hip.so")

    cmd = [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-shared", "-x", "hip"] + srcs + [
        "-o", out,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_dmnist_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-fno-gpu-rdc",
        "-Wno-deprecated-declarations",
    ]
    cmd += [f"-I{p}" for p in torch_inc + [py_inc, ROOT]]
    cmd += [f"-L{torch_lib}", "-ltorch", "-ltorch_cpu", "-ltorch_hip",
            "-lc10", "-lc10_hip", "-ltorch_python",
            "-L/opt/rocm/lib", "-lamdhip64",
            f"-Wl,-rpath,{torch_lib}", "-Wl,-rpath,/opt/rocm/lib"]
    print("[build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    print(f"[build] wrote {out}")


if __name__ == "__main__":
    main()
