"""Native HIP op library loader.

The extension `waternet_amd._C` is built IN-TREE (waternet_amd/_C*.so) by
`python -m waternet_amd.build` (or __graft_entry__.build()) with
hipcc --offload-arch=gfx950. Policy: on a GPU the native kernels are
mandatory — model code calls native_available() and raises if the extension
is missing rather than silently falling back to eager PyTorch.
"""

import importlib

_ext = None
_load_err = None
_tried = False


def _try_load():
    global _ext, _load_err, _tried
    if _tried:
        return
    _tried = True
    try:
        _ext = importlib.import_module("waternet_amd._C")
    except Exception as e:  # noqa: BLE001
        _load_err = repr(e)


def native_available() -> bool:
    _try_load()
    return _ext is not None


def native_load_error():
    _try_load()
    return _load_err


def ext():
    """Return the loaded extension module, raising with the load error if
    unavailable."""
    _try_load()
    if _ext is None:
        raise RuntimeError(
            f"waternet_amd native extension not available: {_load_err}"
        )
    return _ext
