"""Native extension loaders.

``load_des_core()`` — the C++ scalar DES core (CPU; always expected to build).
``load_sim_hip()``  — the batched MI355X HIP engine.  On a machine with a GPU
this must NOT silently fall back: engines fail loudly if the extension is
missing so GPU tests can never pass on an eager-Python substitute.
"""
import importlib
import os


def load_des_core():
    try:
        return importlib.import_module("distributed_cluster_gpus_amd.ops._des_core")
    except ImportError as e:
        raise ImportError(
            "_des_core native extension not built. Run "
            "`python setup.py build_ext --inplace` at the repo root."
        ) from e


def have_des_core() -> bool:
    try:
        load_des_core()
        return True
    except ImportError:
        return False


def load_sim_hip(name: str = "_sim_hip"):
    """Import a gfx950 batched-engine extension ("_sim_hip" = wave-per-replica,
    "_sim_hip_mw" = 8-replicas-per-wave).  torch must be imported first (the
    .so links against libtorch)."""
    import importlib.util
    import torch  # noqa: F401  (symbol provider)
    path = os.path.join(os.path.dirname(__file__), f"{name}.so")
    if not os.path.exists(path):
        raise ImportError(
            f"{name}.so (gfx950 HIP engine) not built. Run "
            "`python -m distributed_cluster_gpus_amd.ops.build_hip` "
            "(requires hipcc; cross-compiles fine without a GPU).")
    import sys
    if name in sys.modules:
        return sys.modules[name]
    spec = importlib.util.spec_from_file_location(name, path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    sys.modules[name] = mod
    return mod


def have_sim_hip() -> bool:
    try:
        load_sim_hip()
        return True
    except ImportError:
        return False
