"""ElasticTrainer / sampler / sharding client behaviors (CPU, no dist)."""

import torch

from dlrover_amd.trainer.elastic import ElasticDistributedSampler, ElasticTrainer


def test_elastic_trainer_accumulation(monkeypatch):
    monkeypatch.setenv("DLROVER_MAX_WORKERS", "4")
    monkeypatch.setenv("WORLD_SIZE", "1")
    model = torch.nn.Linear(4, 4)
    tr = ElasticTrainer(model)
    assert tr.gradient_state.accum_steps == 4
    boundaries = []
    for _ in range(8):
        with tr.step() as sync:
            boundaries.append(sync)
    assert boundaries == [False, False, False, True] * 2
    assert tr.global_step == 2


def test_elastic_trainer_reset_on_world_change(monkeypatch):
    monkeypatch.setenv("DLROVER_MAX_WORKERS", "8")
    monkeypatch.setenv("WORLD_SIZE", "2")
    tr = ElasticTrainer(torch.nn.Linear(2, 2))
    assert tr.gradient_state.accum_steps == 4
    monkeypatch.setenv("WORLD_SIZE", "8")
    tr.reset()
    assert tr.gradient_state.accum_steps == 1


def test_sampler_partitions_disjoint_and_complete():
    data = list(range(20))
    samplers = [
        ElasticDistributedSampler(data, num_replicas=4, rank=r, shuffle=False)
        for r in range(4)
    ]
    seen = []
    for s in samplers:
        seen += list(iter(s))
    assert sorted(seen) == sorted(list(range(20)))


def test_sampler_resume_skips_consumed():
    data = list(range(16))
    s = ElasticDistributedSampler(data, num_replicas=2, rank=0, shuffle=False)
    state = s.state_dict(step=2, batch_size=2)  # consumed 2*2*2=8 samples
    s2 = ElasticDistributedSampler(data, num_replicas=2, rank=0, shuffle=False)
    s2.load_state_dict(state)
    remaining = list(iter(s2))
    assert remaining == [8, 10, 12, 14]
    assert len(s2) == 4


def test_sampler_reshard_after_scale():
    data = list(range(24))
    state = {"epoch": 0, "completed_num": 8}
    # resume with a DIFFERENT world size: remaining samples still disjoint
    samplers = [
        ElasticDistributedSampler(data, num_replicas=3, rank=r, shuffle=False)
        for r in range(3)
    ]
    per_rank = []
    for s in samplers:
        s.load_state_dict(state)
        per_rank.append(list(iter(s)))
    # the remainder (16 % 3 != 0) is PADDED so every rank yields the same
    # count — unequal lengths would hang DDP at epoch end
    assert len({len(x) for x in per_rank}) == 1
    out = [i for x in per_rank for i in x]
    # full coverage of the remaining samples; at most num_replicas-1 dupes
    assert set(out) == set(range(8, 24))
    assert len(out) - len(set(out)) < 3


def test_prepare_gates_optimizer_on_accum_boundary(monkeypatch):
    """Reference-API loop (prepare + step context): optimizer.step() and
    zero_grad() fire only on accumulation boundaries; gradients accumulate
    across the suppressed micro-steps."""
    import torch

    from dlrover_amd.trainer.elastic.trainer import ElasticTrainer

    monkeypatch.setenv("DLROVER_MAX_WORKERS", "4")  # world 1 -> accum 4
    model = torch.nn.Linear(4, 4)
    trainer = ElasticTrainer(model)
    assert trainer.gradient_state.accum_steps == 4

    base_opt = torch.optim.SGD(model.parameters(), lr=0.1)
    sched = torch.optim.lr_scheduler.StepLR(base_opt, step_size=1, gamma=0.5)
    opt, sched = trainer.prepare(base_opt, sched)
    assert opt.param_groups is base_opt.param_groups  # passthrough attrs

    steps_fired = 0
    for micro in range(8):
        with trainer.step() as sync:
            loss = model(torch.randn(2, 4)).sum()
            loss.backward()
            before = [p.clone() for p in model.parameters()]
            opt.step()
            opt.zero_grad()
            changed = any(
                not torch.equal(b, p)
                for b, p in zip(before, model.parameters())
            )
            assert changed == sync
            if sync:
                steps_fired += 1
                sched.step()
            else:
                # grads must survive the gated zero_grad
                assert any(p.grad is not None and p.grad.abs().sum() > 0
                           for p in model.parameters())
    assert steps_fired == 2
    assert trainer.global_step == 2
    assert base_opt.param_groups[0]["lr"] == 0.1 * 0.5 ** 2


def test_elastic_dataloader_applies_versioned_batch_size(monkeypatch):
    """ElasticDataLoader applies a master-pushed batch size once per
    version (ref: dataloader.py + ParallelConfig versioning); an equal or
    older version is a no-op."""
    import torch
    from torch.utils.data import TensorDataset

    from dlrover_amd.common import comm
    from dlrover_amd.trainer.elastic.dataloader import ElasticDataLoader

    ds = TensorDataset(torch.arange(64).float())
    dl = ElasticDataLoader(ds, batch_size=4)

    class FakeClient:
        calls = 0

        def get_paral_config(self):
            FakeClient.calls += 1
            return comm.ParallelConfig(
                dataloader=comm.DataLoaderConfig(batch_size=16, version=1)
            )

    from dlrover_amd.agent import master_client as mc

    monkeypatch.setattr(
        mc.MasterClient, "singleton_instance", classmethod(
            lambda cls: FakeClient()
        ),
    )
    dl.update_batch_size()  # pulls version 1 -> applies 16
    assert dl.batch_size == 16
    assert next(iter(dl))[0].shape[0] == 16
    dl.update_batch_size()  # same version -> no re-apply
    assert FakeClient.calls == 2 and dl.batch_size == 16
    # explicit direct set still works
    dl.update_batch_size(8)
    assert next(iter(dl))[0].shape[0] == 8
    # autotune honors the env gate
    monkeypatch.delenv("DLROVER_AUTO_TUNE", raising=False)
    dl.maybe_autotune()
    assert FakeClient.calls == 2  # gate off: no poll
