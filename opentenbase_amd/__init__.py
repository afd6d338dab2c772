"""opentenbase_amd — MI355X-native (gfx950/CDNA4) query-executor offload for
OpenTenBase: the DataNode SeqScan → HashJoin → HashAggregate hot path plus the
Coordinator shard-merge, as hand-written HIP kernels behind the C-ABI in
include/otbx.h, with RCCL-over-xGMI collectives for the multi-GPU merge.

See DESIGN.md (architecture), SURVEY.md §8 (scope contract), INTEGRATION.md
(how a real OpenTenBase build plugs this in as a CustomScan provider).
"""
__version__ = "0.1"

from ._lib import OtbxError, build  # noqa: F401
