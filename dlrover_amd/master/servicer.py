"""Master RPC dispatch: the single get/report surface every agent talks to.

Parity target: ref dlrover/python/master/servicer.py:89-1140 (MasterServicer
get :152 / report :438 dispatch tables, transport factory :1074). Transports
live in dlrover_amd.utils.transport (tcp/http); this module is pure dispatch.
"""

import time
from typing import Optional

from dlrover_amd.common import comm
from dlrover_amd.common.comm import BaseRequest, BaseResponse
from dlrover_amd.common.constants import RendezvousName
from dlrover_amd.common.log import logger
from dlrover_amd.utils.transport import create_rpc_server


class MasterServicer:
    def __init__(self, master):
        self.master = master  # LocalJobMaster / DistributedJobMaster

    # -- entry ------------------------------------------------------------------

    def handle(self, verb: str, req: BaseRequest) -> BaseResponse:
        msg = req.data
        try:
            if verb == "get":
                data = self._get(req, msg)
            elif verb == "report":
                data = self._report(req, msg)
            else:
                return BaseResponse(success=False, reason=f"unknown verb {verb}")
            return BaseResponse(success=True, data=data)
        except Exception as e:  # noqa: BLE001
            logger.exception("servicer %s(%s) failed", verb, type(msg).__name__)
            return BaseResponse(success=False, reason=repr(e))

    # -- get dispatch --------------------------------------------------------------

    def _get(self, req: BaseRequest, msg) -> Optional[comm.Message]:
        if isinstance(msg, comm.CommWorldRequest):
            mgr = self.master.rdzv_managers[msg.rdzv_name or RendezvousName.TRAINING]
            rnd, group, world = mgr.get_comm_world(msg.node_id)
            return comm.CommWorldResponse(rdzv_round=rnd, group=group, world=world)
        if isinstance(msg, comm.WaitingNodeNumRequest):
            mgr = self.master.rdzv_managers[msg.rdzv_name or RendezvousName.TRAINING]
            return comm.WaitingNodeNumResponse(waiting_num=mgr.num_nodes_waiting())
        if isinstance(msg, comm.KVStoreGetRequest):
            return comm.KeyValuePair(key=msg.key, value=self.master.kv_store.get(msg.key))
        if isinstance(msg, comm.KVStoreMultiGetRequest):
            return comm.KeyValuePairs(kvs=self.master.kv_store.multi_get(msg.keys))
        if isinstance(msg, comm.KVStoreAddRequest):
            return comm.KVStoreAddResponse(value=self.master.kv_store.add(msg.key, msg.amount))
        if isinstance(msg, comm.TaskRequest):
            task = self.master.task_manager.get_task(msg.dataset_name, msg.node_id)
            return task
        if isinstance(msg, comm.ShardCheckpointRequest):
            content = self.master.task_manager.checkpoint_dataset(msg.dataset_name)
            return comm.ShardCheckpoint(dataset_name=msg.dataset_name, content=content)
        if isinstance(msg, comm.NetworkCheckQuery):
            mgr = self.master.rdzv_managers[RendezvousName.NETWORK_CHECK]
            if msg.query == comm.NetworkCheckQuery.QUERY_STRAGGLER:
                return comm.NetworkCheckReply(nodes=mgr.get_stragglers())
            nodes, reason = mgr.check_fault_node()
            return comm.NetworkCheckReply(nodes=nodes, reason=reason)
        if isinstance(msg, comm.RunningNodesRequest):
            nodes = self.master.job_manager.running_nodes()
            return comm.RunningNodes(nodes=nodes)
        if isinstance(msg, comm.ParallelConfigRequest):
            return self.master.paral_config()
        if isinstance(msg, comm.ElasticRunConfigRequest):
            return comm.ElasticRunConfig(configs=self.master.elastic_run_config())
        if isinstance(msg, comm.PreCheckRequest):
            return self.master.pre_check_result(msg.node_id)
        if isinstance(msg, comm.SyncQuery):
            done = self.master.sync_service.is_sync_finished(msg.sync_name)
            return comm.SyncResult(done=done)
        if isinstance(msg, comm.CkptSyncRequest):
            return comm.CkptSyncResponse(
                all_done=self.master.ckpt_sync(msg.node_id, msg.step)
            )
        raise ValueError(f"unhandled get message {type(msg).__name__}")

    # -- report dispatch --------------------------------------------------------------

    def _report(self, req: BaseRequest, msg) -> Optional[comm.Message]:
        if isinstance(msg, comm.JoinRendezvousRequest):
            mgr = self.master.rdzv_managers[msg.rdzv_name or RendezvousName.TRAINING]
            rnd = mgr.join_rendezvous(msg.node_rank, msg.local_world_size)
            self.master.job_manager.on_node_joined(msg.node_rank, msg.node_ip)
            return comm.JoinRendezvousResponse(round=rnd)
        if isinstance(msg, comm.RdzvBlockRequest):
            mgr = self.master.rdzv_managers[msg.rdzv_name or RendezvousName.TRAINING]
            mgr.block_rendezvous(msg.node_rank, msg.blocked)
            return None
        if isinstance(msg, comm.RendezvousParams):
            for mgr in self.master.rdzv_managers.values():
                mgr.update_rdzv_params(
                    msg.min_nodes, msg.max_nodes, msg.waiting_timeout, msg.node_unit
                )
            return None
        if isinstance(msg, comm.KeyValuePair):
            self.master.kv_store.set(msg.key, msg.value)
            return None
        if isinstance(msg, comm.KeyValuePairs):
            self.master.kv_store.multi_set(msg.kvs)
            return None
        if isinstance(msg, comm.KVStoreDeleteRequest):
            self.master.kv_store.delete(msg.key)
            return None
        if isinstance(msg, comm.HeartbeatRequest):
            action = self.master.job_manager.on_heartbeat(
                msg.node_id, msg.node_rank, msg.timestamp or time.time()
            )
            if action is None:
                return comm.HeartbeatResponse()
            cls_name, kwargs = action
            return comm.HeartbeatResponse(action_cls=cls_name, action_kwargs=kwargs)
        if isinstance(msg, comm.NodeEvent):
            self.master.job_manager.on_node_event(msg)
            return None
        if isinstance(msg, comm.NodeFailure):
            self.master.job_manager.on_node_failure(msg)
            return None
        if isinstance(msg, comm.NetworkCheckResult):
            mgr = self.master.rdzv_managers[RendezvousName.NETWORK_CHECK]
            mgr.report_network_check_result(msg.node_id, msg.normal, msg.elapsed_time)
            return None
        if isinstance(msg, comm.DatasetShardParams):
            self.master.task_manager.new_dataset(msg)
            return None
        if isinstance(msg, comm.TaskResult):
            self.master.task_manager.report_task_result(msg)
            return None
        if isinstance(msg, comm.ShardCheckpoint):
            self.master.task_manager.restore_dataset(msg.dataset_name, msg.content)
            return None
        if isinstance(msg, comm.ResourceStats):
            self.master.perf_monitor.report_resource(req.node_id, msg)
            return None
        if isinstance(msg, comm.GlobalStep):
            self.master.perf_monitor.report_global_step(msg.step, msg.timestamp or time.time())
            return None
        if isinstance(msg, comm.SyncJoin):
            self.master.sync_service.join_sync(msg.sync_name, msg.node_id)
            return None
        if isinstance(msg, comm.SyncFinish):
            self.master.sync_service.sync_finished(msg.sync_name)
            return None
        if isinstance(msg, comm.BarrierRequest):
            if msg.notify:
                self.master.sync_service.notify_barrier(msg.barrier_name)
            return comm.SyncResult(done=self.master.sync_service.barrier_reached(msg.barrier_name))
        if isinstance(msg, comm.DiagnosisReportData):
            self.master.diagnosis_manager.collect_data(msg)
            return None
        if isinstance(msg, comm.ModelInfo):
            self.master.perf_monitor.report_model_info(msg)
            return None
        if isinstance(msg, comm.TrainingStatusRequest):
            return comm.TrainingStatusReply(status=self.master.job_manager.training_status())
        raise ValueError(f"unhandled report message {type(msg).__name__}")


def start_master_service(master, service_type: str, port: int):
    servicer = MasterServicer(master)
    server = create_rpc_server(service_type, port, servicer.handle)
    server.start()
    return server
