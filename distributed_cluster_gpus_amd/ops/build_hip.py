"""Build the gfx950 HIP extensions in-tree.

Invokes hipcc directly (cross-compiles fine with no GPU present) and links
against the installed PyTorch-ROCm.  The resulting .so files live inside the
package (they travel to GPU boxes with the source snapshot; a JIT cache under
~/.cache would not).

Usage:  python -m distributed_cluster_gpus_amd.ops.build_hip
"""
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc", "hip")

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch  # noqa
    import torch.utils.cpp_extension as ce
    inc = ce.include_paths()
    lib = ce.library_paths()
    return inc, lib


def build_extension(name: str, sources, verbose=True, extra_defs=()) -> str:
    inc, lib = torch_paths()
    import sysconfig
    py_inc = sysconfig.get_paths()["include"]
    out = os.path.join(HERE, f"{name}.so")
    cmd = ["hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
           "-shared", f"-DTORCH_EXTENSION_NAME={name}", "-DUSE_ROCM",
           "-DGLOG_USE_GLOG_EXPORT",
           "-Wno-unused-result"] + list(extra_defs)
    for i in inc:
        cmd.append(f"-I{i}")
    cmd.append(f"-I{py_inc}")
    cmd.append(f"-I{CSRC}")
    cmd += [os.path.join(CSRC, s) for s in sources]
    for l in lib:
        cmd.append(f"-L{l}")
    cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
            "-ltorch_hip", "-lc10_hip", "-lamdhip64", "-o", out]
    if verbose:
        print("[build_hip]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return out


def build_all(verbose=True):
    return [
        build_extension("_sim_hip", ["replica_engine.hip"], verbose=verbose),
        # multi-replica-per-wave variant: 8 replicas x 8 lanes per wavefront
        build_extension("_sim_hip_mw", ["replica_engine.hip"], verbose=verbose,
                        extra_defs=["-DDCG_SUBWAVE=8"]),
    ]


if __name__ == "__main__":
    for so in build_all():
        print("built", so)
