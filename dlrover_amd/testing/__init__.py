"""Multi-node-without-a-cluster test harness.

Parity target: ref dlrover/python/testing/ (master_setup.py pattern: spin up
a REAL master subprocess, connect fake agents over real RPC, drive failure
scenarios through production code paths — SURVEY.md §4).

Used by tests/test_elastic_scale_e2e.py etc.; exported here so downstream
users can script their own chaos scenarios.
"""

from dlrover_amd.testing.harness import (  # noqa: F401
    FakeAgent,
    MasterProcess,
    run_scenario,
)
