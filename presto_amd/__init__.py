"""presto_amd — MI355X-native implementation of Presto's per-Page operator
hot path (ScanFilterAndProject / HashAggregation / HashBuilder+LookupJoin /
TopN / PartitionedOutput) as hand-written CDNA4 HIP kernels behind the C-ABI
of include/presto_gpu.h.

This package is the host-side mirror of the reference's Operator surface
(presto-main-base/.../operator/Operator.java:20-102): the Operator class
exposes needs_input/add_input/get_output/finish/is_finished with the same
meaning, and the pipelines in presto_amd.pipelines drive them exactly like
Driver.processInternal (operator/Driver.java:402-475).

There is NO CPU fallback: importing works anywhere (so symbol tests run
without a GPU) but creating any operator requires an AMD gfx950 GPU and
fails loudly otherwise.
"""
from .engine import (  # noqa: F401
    lib, Operator, Page, Varbin, DictVarbin, DeviceVarbin,
    PlanFilterProject,
    PlanHashAggSmall,
    PlanHashBuild, PlanLookupJoin, PlanTopN, PlanPartition, PlanGroupBy,
    Pred, Proj, Agg,
    CMP_LT, CMP_LE, CMP_GT, CMP_GE, CMP_EQ, CMP_NE, CMP_CONTAINS,
    CMP_PREFIX, CMP_CONTAINS2, CMP_NOT_CONTAINS2,
    PROJ_IDENT, PROJ_DISC_PRICE, PROJ_CHARGE, PROJ_MUL, PROJ_DIV,
    PROJ_KEYSHL, PROJ_SHR, PROJ_SUBDIV, PROJ_KEYSHL_DIV,
    AGG_COUNT, AGG_SUM_F64, AGG_SUM_DEC, AGG_SUM_I64, AGG_MIN, AGG_MAX,
    OP_FILTER_PROJECT, OP_HASH_AGG_SMALL, OP_HASH_BUILD, OP_LOOKUP_JOIN,
    OP_TOPN, OP_PARTITION, OP_GROUPBY_MULTI,
)
from . import pipelines  # noqa: F401
