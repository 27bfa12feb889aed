# Importing torch first puts libc10/libtorch on the loader path for the
# pybind11 extension built with torch's toolchain.
import torch  # noqa: F401

from . import _dtx_native  # noqa: F401
