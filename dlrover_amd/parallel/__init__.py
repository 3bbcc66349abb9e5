from dlrover_amd.parallel.pgroups import ParallelDims, ParallelGroups  # noqa: F401
