# baikaldb_amd — MI355X-native implementation of BaikalDB's OLAP execution
# hot path (scan -> filter -> hash-aggregate / top-N sort), built from scratch
# for gfx950/CDNA4 behind the reference's ExecNode plugin surface.
#
# The compute path is the in-tree HIP extension baikaldb_amd/libbkgpu.so
# (C-ABI: include/bkgpu.h). There is NO CPU fallback: on a GPU machine the
# engine refuses to run without the native extension (the CPU oracle under
# oracle/ is test infrastructure only).
from .engine import GpuEngine, GpuTable, NativeEngineMissing  # noqa: F401
from .plan import QueryPlan  # noqa: F401
