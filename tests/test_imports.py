"""Every module in the package imports cleanly (catches dangling imports
in rarely-exercised corners; lightning is import-guarded)."""

import importlib
import os
import pkgutil

import neuronx_distributed_amd as nxd


def test_all_modules_import():
    root = os.path.dirname(nxd.__file__)
    failed = []
    for mod in pkgutil.walk_packages([root], prefix="neuronx_distributed_amd."):
        name = mod.name
        if ".ops.csrc" in name or name.endswith("libnxd_ops"):
            continue  # ctypes-loaded HIP library, not a python module
        try:
            importlib.import_module(name)
        except Exception as e:  # pragma: no cover
            failed.append((name, repr(e)))
    assert not failed, failed
