#!/usr/bin/env python3
"""Build entry for the in-tree gfx950 HIP extension.

`python setup.py build_ext --inplace` (the command ops/flash.py's error
message names) compiles ops/hip/{fa_kernels.hip,bindings.cpp} with hipcc
for gfx950 via torch.utils.cpp_extension and leaves _tree_attn_hip.so
next to the sources — same build __graft_entry__.build() runs. Package
installation metadata lives in pyproject.toml.
"""

import sys

if __name__ == "__main__":
    if "build_ext" in sys.argv:
        sys.path.insert(0, ".")
        from tree_attention_torch_amd.ops.hip.build import build

        print(build())
    else:
        from setuptools import setup

        setup()
