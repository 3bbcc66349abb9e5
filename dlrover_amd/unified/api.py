"""Unified MPMD job API.

Parity target: ref dlrover/python/unified/api/builder/base.py:363-631
(DLJobBuilder: .train()/.role() chains, collocation, submit -> PrimeMaster).
The reference runs roles as Ray actors; Ray is not in the MI355X image, so
the execution backend is local processes (unified/master.py) with the same
role/graph/failover semantics — the API shape is preserved so user job
definitions carry over.
"""

from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional


@dataclass
class WorkloadDesc:
    """One role's spec (ref: unified/common/workload_desc.py)."""

    name: str = "trainer"
    total: int = 1
    per_group: int = 1
    entry_func: Optional[Callable] = None
    entry_module: str = ""
    entry_args: tuple = ()
    env: Dict[str, str] = field(default_factory=dict)
    max_restarts: int = 1
    resource: Dict[str, float] = field(default_factory=dict)


@dataclass
class DLJob:
    name: str = "dljob"
    roles: Dict[str, WorkloadDesc] = field(default_factory=dict)
    collocations: List[List[str]] = field(default_factory=list)
    node_unit: int = 1

    def submit(self, blocking: bool = True):
        from dlrover_amd.unified.master import PrimeMaster

        master = PrimeMaster(self)
        master.prepare()
        master.start()
        if blocking:
            master.wait()
        return master


class _RoleBuilder:
    def __init__(self, parent: "DLJobBuilder", desc: WorkloadDesc):
        self._parent = parent
        self._desc = desc

    def run(self, fn: Callable, *args) -> "_RoleBuilder":
        self._desc.entry_func = fn
        self._desc.entry_args = args
        return self

    def run_module(self, module: str, *args) -> "_RoleBuilder":
        self._desc.entry_module = module
        self._desc.entry_args = args
        return self

    def total(self, n: int) -> "_RoleBuilder":
        self._desc.total = n
        return self

    def per_group(self, n: int) -> "_RoleBuilder":
        self._desc.per_group = n
        return self

    def resource(self, **kw) -> "_RoleBuilder":
        self._desc.resource.update(kw)
        return self

    def env(self, **kw) -> "_RoleBuilder":
        self._desc.env.update({k: str(v) for k, v in kw.items()})
        return self

    def max_restarts(self, n: int) -> "_RoleBuilder":
        self._desc.max_restarts = n
        return self

    # chain back
    def role(self, name: str) -> "_RoleBuilder":
        return self._parent.role(name)

    def __getattr__(self, name):
        # any other chain method (train, actor, rollout, ...) belongs to the
        # job builder: delegate so role chains compose naturally
        return getattr(self._parent, name)

    def with_collocation(self, *names: str) -> "DLJobBuilder":
        return self._parent.with_collocation(*names)

    def node_unit(self, n: int) -> "DLJobBuilder":
        return self._parent.node_unit(n)

    def build(self) -> DLJob:
        return self._parent.build()


class DLJobBuilder:
    """DLJobBuilder().train(4).run(fn).build().submit()"""

    def __init__(self, name: str = "dljob"):
        self._job = DLJob(name=name)

    def role(self, name: str) -> _RoleBuilder:
        desc = self._job.roles.setdefault(name, WorkloadDesc(name=name))
        return _RoleBuilder(self, desc)

    def train(self, total: int = 1, per_group: int = 0) -> _RoleBuilder:
        rb = self.role("trainer").total(total)
        if per_group:
            rb.per_group(per_group)
        return rb

    def with_collocation(self, *names: str) -> "DLJobBuilder":
        self._job.collocations.append(list(names))
        return self

    def node_unit(self, n: int) -> "DLJobBuilder":
        self._job.node_unit = n
        return self

    def build(self) -> DLJob:
        if not self._job.roles:
            raise ValueError("job has no roles")
        for r in self._job.roles.values():
            if r.entry_func is None and not r.entry_module:
                raise ValueError(f"role {r.name} has no entrypoint")
        self._validate(self._job)
        return self._job

    def _validate(self, job: DLJob) -> None:  # extension hook
        pass


class RLJobBuilder(DLJobBuilder):
    """Reinforcement-learning job builder (ref: api/builder/rl.py:149):
    the RL role vocabulary (trainer/actor/reference/reward/critic/rollout)
    with per-role helpers and build-time validation — 'actor' is mandatory,
    unknown roles are rejected.

        job = (RLJobBuilder()
               .trainer().run(train_fn)
               .actor(8).run(actor_fn)
               .rollout(4).run(rollout_fn)
               .with_collocation("actor", "rollout")
               .build())
    """

    TRAINER_ROLE = "trainer"
    ACTOR_ROLE = "actor"
    REF_ROLE = "reference"
    REW_ROLE = "reward"
    CRITIC_ROLE = "critic"
    ROLLOUT_ROLE = "rollout"
    ROLES = [TRAINER_ROLE, ACTOR_ROLE, REF_ROLE, REW_ROLE, CRITIC_ROLE,
             ROLLOUT_ROLE]

    def trainer(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.TRAINER_ROLE).total(total)

    def actor(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.ACTOR_ROLE).total(total)

    def reference(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.REF_ROLE).total(total)

    def reward(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.REW_ROLE).total(total)

    def critic(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.CRITIC_ROLE).total(total)

    def rollout(self, total: int = 1) -> _RoleBuilder:
        return self.role(self.ROLLOUT_ROLE).total(total)

    def _validate(self, job: DLJob) -> None:
        if self.ACTOR_ROLE not in job.roles:
            raise ValueError("'actor' must be configured for an RL job")
        for role in job.roles:
            if role not in self.ROLES:
                raise ValueError(
                    f"invalid role {role!r} for an RL job; supported: "
                    f"{self.ROLES}"
                )
