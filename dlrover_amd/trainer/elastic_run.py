"""``dlrover-run`` — the elastic launcher (torchrun superset).

Parity target: ref dlrover/trainer/torch/elastic_run.py:1-643 — argument
surface (--network-check, --node-unit, --max-restarts, --standalone, ...),
standalone local-master spawn (:326), master pre-check wait (:295), then
launch_agent.

Usage:
    dlrover-run --standalone --nproc-per-node 8 train.py ARGS...
    dlrover-run --nnodes 2:4 --nproc-per-node 8 \
        --rdzv-endpoint $MASTER_ADDR:24666 train.py ARGS...
"""

import argparse
import os
import subprocess
import sys
import tempfile
import time
import uuid
from typing import List, Optional, Tuple

from dlrover_amd.agent.master_client import MasterClient
from dlrover_amd.agent.training import ElasticLaunchConfig, launch_agent
from dlrover_amd.common import comm
from dlrover_amd.common.constants import CommServiceType, NodeEnv
from dlrover_amd.common.log import logger
from dlrover_amd.utils.transport import wait_for_server


def parse_nnodes(val: str) -> Tuple[int, int]:
    if ":" in val:
        lo, hi = val.split(":")
        lo_i, hi_i = int(lo), int(hi)
        if lo_i < 1 or hi_i < lo_i:
            raise ValueError(f"--nnodes {val}: need 1 <= MIN <= MAX")
        return lo_i, hi_i
    n = int(val)
    if n < 1:
        raise ValueError(f"--nnodes {val}: need >= 1 node")
    return n, n


def parse_args(argv: Optional[List[str]] = None):
    p = argparse.ArgumentParser(
        "dlrover-run", description="MI355X-native elastic training launcher"
    )
    p.add_argument("--nnodes", default="1", help="N or MIN:MAX for elasticity")
    p.add_argument("--nproc-per-node", "--nproc_per_node", type=int, default=0,
                   help="workers per node (0 = one per visible GPU)")
    p.add_argument("--node-rank", "--node_rank", type=int,
                   default=int(os.getenv(NodeEnv.NODE_RANK, os.getenv(NodeEnv.NODE_ID, "0"))))
    p.add_argument("--max-restarts", "--max_restarts", type=int, default=3)
    p.add_argument("--monitor-interval", type=float, default=5.0)
    p.add_argument("--rdzv-endpoint", "--rdzv_endpoint", default="",
                   help="master addr host:port (or DLROVER_MASTER_ADDR env)")
    p.add_argument("--standalone", action="store_true",
                   help="spawn a local job master in-process")
    p.add_argument("--rdzv-conf", "--rdzv_conf", default="",
                   help="k=v,... extra rendezvous config (pet compat)")
    p.add_argument("--join-timeout", type=float, default=600.0)
    p.add_argument("--waiting-timeout", type=float, default=60.0)
    p.add_argument("--network-check", "--network_check", action="store_true",
                   help="run matmul+allreduce probes before training")
    p.add_argument("--comm-perf-test", "--comm_perf_test", action="store_true")
    p.add_argument("--precheck", type=int, default=0, choices=[0, 1, 2],
                   help="0=off, 1=comm probes, 2=probes + straggler "
                        "exclusion (ref precheck levels)")
    p.add_argument("--exclude-straggler", "--exclude_straggler",
                   action="store_true",
                   help="refuse to train on a straggler node (the probe "
                        "raises instead of observing)")
    p.add_argument("--training-port", "--training_port", type=int, default=0,
                   help="accepted for reference-CLI compatibility (NPU port "
                        "sync; not needed on MI355X)")
    p.add_argument("--membind-policy", "--membind_policy", default="",
                   help="NUMA memory policy hint for --numa-affinity")
    p.add_argument("--node-unit", "--node_unit", type=int, default=1,
                   help="world size must be a multiple of this")
    p.add_argument("--auto-config", action="store_true",
                   help="fill nproc-per-node from the visible GPU count "
                        "(ref auto_configure_params)")
    p.add_argument("--auto-tunning", action="store_true",
                   help="let workers poll master-generated hyperparameter "
                        "suggestions (versioned ParallelConfig)")
    p.add_argument("--numa-affinity", action="store_true")
    p.add_argument("--accelerator", default="amd.com/gpu")
    p.add_argument("--save-at-breakpoint", "--save_at_breakpoint",
                   action="store_true",
                   help="accepted for reference-CLI compatibility: the agent "
                        "ALWAYS persists the latest shm checkpoint to "
                        "storage on worker failure/SIGTERM in this build "
                        "(set DLROVER_NO_BREAKPOINT_SAVE=1 to disable)")
    p.add_argument("--hiptimer", action="store_true",
                   help="LD_PRELOAD the hiptimer profiler into workers "
                        "(kernel/GEMM/RCCL timing + hang detection)")
    p.add_argument("--checkpoint-dir", default="/tmp/dlrover_amd_ckpt")
    p.add_argument("--log-dir", default=None)
    p.add_argument("--redirects", default="0",
                   help="redirect worker std streams to files "
                        "(0|1|2|3 = none|stdout|stderr|both, torchrun-style)")
    p.add_argument("--tee", default="0",
                   help="tee worker std streams to console AND files")
    p.add_argument("--run-id", "--run_id", default="",
                   help="rendezvous run id (defaults to the job name)")
    p.add_argument("--service-type", default=CommServiceType.TCP)
    # torchrun flags accepted for drop-in compatibility; rendezvous is the
    # dlrover master here, so the c10d knobs are intentionally ignored
    for kebab in ("rdzv-backend", "rdzv-id", "start-method", "local-addr",
                  "master-addr", "master-port"):
        p.add_argument(f"--{kebab}", f"--{kebab.replace('-', '_')}",
                       default=None, help=argparse.SUPPRESS)
    p.add_argument("--no-python", "--no_python", action="store_true",
                   help="run the script directly (torchrun compat; implied "
                        "for non-.py scripts)")
    p.add_argument("training_script", help="training program (.py or executable)")
    p.add_argument("training_script_args", nargs=argparse.REMAINDER)
    args = p.parse_args(argv)
    ignored = [f for f in ("rdzv_backend", "rdzv_id", "start_method",
                           "local_addr", "master_addr", "master_port")
               if getattr(args, f, None)]
    if ignored:
        logger.info(
            "torchrun-compat flags ignored (the dlrover master handles "
            "rendezvous): %s", ", ".join(ignored),
        )
    return args


def _launch_local_master(service_type: str) -> Tuple[subprocess.Popen, str]:
    """Spawn a standalone local master and wait for its port
    (ref: _launch_dlrover_local_master :326)."""
    port_file = os.path.join(
        tempfile.gettempdir(), f"dlrover_master_{uuid.uuid4().hex[:8]}.port"
    )
    proc = subprocess.Popen(
        [
            sys.executable,
            "-m",
            "dlrover_amd.master.main",
            "--platform",
            "local",
            "--port",
            "0",
            "--service_type",
            service_type,
            "--port_file",
            port_file,
        ],
        env=dict(os.environ),
    )
    deadline = time.time() + 60
    while time.time() < deadline:
        if os.path.exists(port_file):
            with open(port_file) as f:
                content = f.read().strip()
            if content:
                addr = f"127.0.0.1:{content}"
                if wait_for_server(addr, timeout=30, service_type=service_type):
                    return proc, addr
        if proc.poll() is not None:
            raise RuntimeError("local master exited during startup")
        time.sleep(0.2)
    proc.terminate()
    raise RuntimeError("local master did not start in time")


def wait_pre_check(client: MasterClient, timeout: float = 600):
    """Poll master pre-check (ref: wait_pre_check :295)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        resp = client.get_pre_check_result()
        if resp.status == comm.PreCheckResponse.PASS:
            return
        if resp.status == comm.PreCheckResponse.FAIL:
            raise RuntimeError(f"master pre-check failed: {resp.reason}")
        time.sleep(5)
    raise TimeoutError("master pre-check did not complete")


def run(args) -> int:
    master_proc = None
    if args.standalone and args.node_rank == 0 and not args.rdzv_endpoint:
        master_proc, master_addr = _launch_local_master(args.service_type)
        logger.info("standalone master at %s", master_addr)
    else:
        master_addr = args.rdzv_endpoint or os.getenv(NodeEnv.MASTER_ADDR, "")
        if not master_addr:
            raise SystemExit(
                "no master: pass --standalone or --rdzv-endpoint host:port"
            )
        if not wait_for_server(master_addr, timeout=120, service_type=args.service_type):
            raise SystemExit(f"master {master_addr} unreachable")

    # worker processes must be able to import dlrover_amd regardless of the
    # script's location (the package runs in-tree, not from site-packages)
    import dlrover_amd

    pkg_root = os.path.dirname(os.path.dirname(os.path.abspath(dlrover_amd.__file__)))
    existing = os.environ.get("PYTHONPATH", "")
    if pkg_root not in existing.split(os.pathsep):
        os.environ["PYTHONPATH"] = (
            f"{pkg_root}{os.pathsep}{existing}" if existing else pkg_root
        )

    os.environ[NodeEnv.MASTER_ADDR] = master_addr
    os.environ[NodeEnv.MASTER_SERVICE_TYPE] = args.service_type
    os.environ[NodeEnv.NODE_ID] = str(args.node_rank)
    os.environ[NodeEnv.NODE_RANK] = str(args.node_rank)
    os.environ.setdefault("ELASTIC_JOB_NAME", f"job-{uuid.uuid4().hex[:6]}")
    MasterClient.reset()
    client = MasterClient.singleton_instance()

    # master-pushed config overrides (ref: elastic_run.py:438-477)
    try:
        overrides = client.get_elastic_run_config()
        if overrides:
            logger.info("master config overrides: %s", overrides)
            if "network_check" in overrides:
                args.network_check = overrides["network_check"] == "true"
    except Exception:  # noqa: BLE001
        pass

    if args.hiptimer:
        os.environ["DLROVER_HIPTIMER"] = "1"

    # announce this agent BEFORE the pre-check gate so master-side checks
    # (e.g. the min_nodes operator) can count arrivals
    try:
        client.report_node_event("ADDED", "agent started", args.node_rank)
    except Exception:  # noqa: BLE001
        pass

    wait_pre_check(client)

    # --precheck levels map onto the probe flags (ref precheck semantics)
    if args.precheck >= 1:
        args.network_check = True
    if args.precheck >= 2:
        args.exclude_straggler = True
    if args.membind_policy:
        os.environ["DLROVER_MEMBIND_POLICY"] = args.membind_policy

    min_nodes, max_nodes = parse_nnodes(args.nnodes)
    if args.auto_config and args.nproc_per_node <= 0:
        args.nproc_per_node = 0  # auto_configure() fills from device count
    if args.auto_tunning:
        os.environ["DLROVER_AUTO_TUNE"] = "1"
    config = ElasticLaunchConfig(
        min_nodes=min_nodes,
        max_nodes=max_nodes,
        nproc_per_node=args.nproc_per_node,
        node_rank=args.node_rank,
        max_restarts=args.max_restarts,
        monitor_interval=args.monitor_interval,
        rdzv_timeout=args.join_timeout,
        waiting_timeout=args.waiting_timeout,
        node_unit=args.node_unit,
        network_check=args.network_check,
        comm_perf_test=args.comm_perf_test,
        exclude_straggler=args.exclude_straggler,
        checkpoint_dir=args.checkpoint_dir,
        log_dir=args.log_dir,
        numa_affinity=args.numa_affinity,
        redirects=args.redirects,
        tee=args.tee,
        run_id=args.run_id or os.getenv("ELASTIC_JOB_NAME", "dlrover"),
    )

    config.auto_configure()
    script = args.training_script
    script_args = list(args.training_script_args)
    if script.endswith(".py") and not getattr(args, "no_python", False):
        entrypoint: object = sys.executable
        script_args = ["-u", script] + script_args
    else:
        entrypoint = script

    try:
        launch_agent(config, entrypoint, script_args)
        return 0
    except Exception:
        logger.exception("dlrover-run failed")
        return 1
    finally:
        if master_proc is not None:
            master_proc.terminate()
            try:
                master_proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                master_proc.kill()


def main(argv: Optional[List[str]] = None) -> int:
    return run(parse_args(argv))


if __name__ == "__main__":
    sys.exit(main())
