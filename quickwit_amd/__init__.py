"""quickwit_amd — MI355X-native Quickwit leaf-search hot path.

The product is the C-ABI shared library `libquickwit_amd.so` (C++ host + HIP
gfx950 kernels; see include/quickwit_amd.h and DESIGN.md). This package holds
the Python harness around it: ctypes bindings (`quickwit_amd.api`), the
protobuf codec for the search.proto messages (`quickwit_amd.proto`), the QWA1
split writer/generator (`quickwit_amd.splitgen`) and the cross-rank merge used
by bench.py (`quickwit_amd.merge`).
"""

__version__ = "0.1"
