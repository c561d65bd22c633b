"""galaxysql_amd — MI355X-native operator pack for the PolarDB-X CN MPP hot
path (ParallelHashJoinExec -> HashAggExec -> LocalExchanger PARTITION),
built from scratch behind the reference's operator API.

Product compute path: galaxysql_amd/csrc (HIP/gfx950 kernels behind the
C-ABI in include/gxop.h). The host layer here mirrors the reference's
Executor/ConsumerExecutor/Chunk surface so the physical plan's operators
drop in unchanged (see INTEGRATION.md for the JNI binding the real CN would
add).
"""

from . import chunk, abi, operators  # noqa: F401

__version__ = "0.1.0"
