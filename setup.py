#!/usr/bin/env python3
"""`python setup.py build_ext --inplace` -> dist_mnist_amd/_C.so (gfx950).

Thin wrapper over tools/build_ext.py (a single explicit hipcc invocation; no
hipify pass — the sources under native/ are native HIP for CDNA4)."""

import sys

if __name__ == "__main__":
    if "build_ext" in sys.argv or "build" in sys.argv:
        from tools.build_ext import build

        build()
        print("built dist_mnist_amd/_C.so for gfx950")
    else:
        print("usage: python setup.py build_ext --inplace")
        sys.exit(1)
