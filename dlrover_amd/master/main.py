"""Master entry point: ``python -m dlrover_amd.master.main --platform local
--port 0 ...`` (ref: dlrover/python/master/main.py:48-150)."""

import argparse
import os
import sys

from dlrover_amd.common.constants import CommServiceType, PlatformType
from dlrover_amd.common.log import logger


def parse_args(argv=None):
    p = argparse.ArgumentParser("dlrover_amd master")
    p.add_argument("--platform", default=PlatformType.LOCAL,
                   choices=[PlatformType.LOCAL, PlatformType.KUBERNETES])
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--service_type", default=CommServiceType.TCP,
                   choices=[CommServiceType.TCP, CommServiceType.HTTP,
                            CommServiceType.GRPC])
    p.add_argument("--job_name", default=os.getenv("ELASTIC_JOB_NAME", "dlrover-job"))
    p.add_argument("--namespace", default="default")
    p.add_argument("--port_file", default="",
                   help="write the bound port here (standalone launcher handshake)")
    p.add_argument("--enable_dashboard", action="store_true")
    p.add_argument("--dashboard_port", type=int, default=0)
    return p.parse_args(argv)


def run(args) -> int:
    if args.platform == PlatformType.KUBERNETES:
        from dlrover_amd.master.job_master import DistributedJobMaster
        from dlrover_amd.master.scaler.pod_scaler import PodScaler
        from dlrover_amd.master.watcher.k8s_watcher import (
            ElasticJobWatcher,
            PodWatcher,
            ScalePlanWatcher,
        )

        scaler = PodScaler(args.job_name, args.namespace)
        watcher = PodWatcher(args.job_name, args.namespace)
        job_watcher = ElasticJobWatcher(args.job_name, args.namespace)
        scaleplan_watcher = ScalePlanWatcher(args.job_name, args.namespace)
        master = DistributedJobMaster(
            scaler=scaler, watcher=watcher, job_watcher=job_watcher,
            scaleplan_watcher=scaleplan_watcher,
            port=args.port, service_type=args.service_type,
        )
    else:
        from dlrover_amd.master.job_master import LocalJobMaster

        master = LocalJobMaster(port=args.port, service_type=args.service_type)
    master.prepare()
    dashboard = None
    if args.enable_dashboard:
        from dlrover_amd.master.dashboard import Dashboard

        dashboard = Dashboard(master, port=args.dashboard_port).start()
    if args.port_file:
        with open(args.port_file, "w") as f:
            f.write(str(master.port))
    logger.info("master for job %s listening on %s", args.job_name, master.port)
    try:
        return master.run()
    finally:
        if dashboard is not None:
            dashboard.stop()
        master.stop()


def main(argv=None) -> int:
    return run(parse_args(argv))


if __name__ == "__main__":
    sys.exit(main())
