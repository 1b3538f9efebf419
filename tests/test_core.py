"""Core representation tests: HParams exclusivity, Task checkpointing,
iterator resume, Strategy validation, library round trip."""

import os

import pytest
import torch

import saturn_amd
from saturn_amd import BaseTechnique, HParams, Strategy, Task, register, retrieve, deregister
from saturn_amd.models import get_mlp_dataloader, get_mlp_model, mse_loss


def test_hparams_exclusivity():
    with pytest.raises(ValueError):
        HParams(lr=1e-3)
    with pytest.raises(ValueError):
        HParams(lr=1e-3, epochs=1, batch_count=5)
    assert HParams(lr=1e-3, epochs=2).epochs == 2
    assert HParams(lr=1e-3, batch_count=7).batch_count == 7


def test_strategy_validation():
    with pytest.raises(ValueError):
        Strategy(None, 0)
    with pytest.raises(ValueError):
        Strategy(None, 1.5)
    s = Strategy(None, 2, {"x": 1}, 10.0, batch_time=0.1)
    assert not s.feasible  # executor is None
    s2 = Strategy(object, 2, {"x": 1}, 10.0)
    assert s2.feasible


def test_task_lazy_epoch_length(save_dir):
    calls = []

    def dl():
        calls.append(1)
        return [1, 2, 3, 4]

    t = Task(get_mlp_model, dl, mse_loss, HParams(lr=1e-3, batch_count=10),
             save_dir=save_dir)
    assert not calls  # not instantiated in __init__ (reference quirk #6 fixed)
    assert t.epoch_length == 4
    assert calls
    assert t.total_batches == 10


def test_task_epochs_total_batches(save_dir):
    t = Task(get_mlp_model, lambda: [0] * 5, mse_loss,
             HParams(lr=1e-3, epochs=3), save_dir=save_dir)
    assert t.total_batches == 15


def test_task_checkpoint_roundtrip(save_dir):
    t = Task(get_mlp_model, get_mlp_dataloader, mse_loss,
             HParams(lr=1e-3, batch_count=4), name="ck", save_dir=save_dir)
    assert not t.has_ckpt()
    m = t.get_model(fresh=True)
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    # run a step so optimizer has state-like content and weights change
    x, y = next(iter(get_mlp_dataloader()))
    mse_loss(m(x), y).backward()
    opt.step()
    t.save_checkpoint(m, opt)
    assert t.has_ckpt()
    assert os.path.isfile(os.path.join(save_dir, "ck.pt"))
    m2 = t.get_model()
    for a, b in zip(m.state_dict().values(), m2.state_dict().values()):
        assert torch.equal(a, b)
    ck = t.load_checkpoint()
    assert ck["optimizer"] is not None


def test_task_iterator_resume(save_dir):
    t = Task(get_mlp_model, lambda: iter(range(10)), mse_loss,
             HParams(lr=1e-3, batch_count=10), save_dir=save_dir)
    t._epoch_length = 10
    t.current_batch = 3
    it = t.get_iterator()
    assert next(it) == 3
    t.reconfigure(4)
    assert t.current_batch == 7
    t.reconfigure(5)  # wraps the epoch
    assert t.current_batch == 2


def test_transformer_hint_validation(save_dir):
    with pytest.raises(ValueError):
        Task(get_mlp_model, get_mlp_dataloader, mse_loss,
             HParams(lr=1e-3, batch_count=1),
             hints={"is_transformer": True}, save_dir=save_dir)
    t = Task(get_mlp_model, get_mlp_dataloader, mse_loss,
             HParams(lr=1e-3, batch_count=1),
             hints={"is_transformer": True, "transformer_cls": {torch.nn.Linear}},
             save_dir=save_dir)
    assert t.hints["is_transformer"]


class DummyTech(BaseTechnique):
    name = "dummy"

    @staticmethod
    def execute(task, gpus, tid, batch_count):
        return None

    @staticmethod
    def search(task, gpus, tid):
        return {"k": 1}, 0.5


def test_library_roundtrip(library_path):
    register("dummy", DummyTech)
    assert os.path.isfile(os.path.join(library_path, "dummy.udp"))
    cls = retrieve("dummy")
    assert cls.name == "dummy"
    assert cls.search(None, [0], 0) == ({"k": 1}, 0.5)
    all_techs = retrieve()
    assert len(all_techs) == 1
    deregister("dummy")
    assert not os.listdir(library_path)


def test_library_rejects_non_technique(library_path):
    with pytest.raises(RuntimeError):
        register("bad", int)


def test_library_deregister_list(library_path):
    register("a", DummyTech)
    register("b", DummyTech)
    deregister(["a", "b"])  # reference's list path was buggy (library.py:45-47)
    assert not os.listdir(library_path)


def test_public_api_surface():
    for sym in ["Task", "HParams", "Strategy", "Techniques", "BaseTechnique",
                "register", "deregister", "retrieve", "search", "orchestrate",
                "solve", "Plan"]:
        assert hasattr(saturn_amd, sym)


def test_text_dataloader_cache_roundtrip(tmp_path):
    from saturn_amd.models.data import load_text_dataset, make_text_dataloader

    p = tmp_path / "corpus.txt"
    p.write_text("hello world, this is a tiny corpus for windowing. " * 50)
    w1 = load_text_dataset(str(p), context_length=64)
    assert w1.shape[1] == 64 and w1.shape[0] > 10
    assert (tmp_path / "corpus.txt.ctx64.npz").exists()
    w2 = load_text_dataset(str(p), context_length=64)  # cache hit
    assert torch.equal(w1, w2)
    dl = make_text_dataloader(str(p), batch_size=4, context_length=64)()
    x, y = next(iter(dl))
    assert x.shape == (4, 64) and torch.equal(x, y)


def test_hparams_validation_errors():
    import pytest as _pytest

    with _pytest.raises(Exception):
        HParams(lr=1e-3)  # neither epochs nor batch_count
    with _pytest.raises(Exception):
        HParams(lr=1e-3, epochs=1, batch_count=5)  # both


def test_strategy_validation():
    import pytest as _pytest

    with _pytest.raises(ValueError):
        Strategy(None, 0)  # gpu count must be > 0
    with _pytest.raises(ValueError):
        Strategy(None, -2)
    # feasible requires a real executor AND params (the reference's DDP
    # was never selectable because it returned params=None — quirk #2)
    s = Strategy(None, 2, runtime=10.0, batch_time=1.0)
    assert not s.feasible
    s2 = Strategy(object, 2, {"p": 1}, runtime=10.0, batch_time=1.0)
    assert s2.feasible


def test_solve_empty_task_list():
    from saturn_amd.solver import solve

    plan = solve([], n_gpus=4, timeout=5)
    assert plan.task_names == [] and plan.makespan == 0.0


def test_skip_ckpt_env_and_delete(tmp_path, monkeypatch):
    """SATURN_SKIP_CKPT=1 makes save_checkpoint a no-op (pure-makespan
    benchmarking on small scratch disks); delete_checkpoint removes the
    ckpt and any per-rank shard files."""
    import torch

    from saturn_amd import HParams, Task

    t = Task(
        lambda: torch.nn.Linear(2, 2),
        lambda: [0],
        lambda a, b: None,
        HParams(lr=1e-3, batch_count=1),
        name="ckpt_env",
        save_dir=str(tmp_path),
    )
    m = torch.nn.Linear(2, 2)
    monkeypatch.setenv("SATURN_SKIP_CKPT", "1")
    t.save_checkpoint(m)
    assert not t.has_ckpt()
    monkeypatch.delenv("SATURN_SKIP_CKPT")
    t.save_checkpoint(m)
    assert t.has_ckpt()
    shard = tmp_path / "ckpt_env.optshard.w2.r0.pt"
    shard.write_bytes(b"x")
    t.delete_checkpoint()
    assert not t.has_ckpt() and not shard.exists()


def test_port_pool_disjoint_and_env(monkeypatch):
    """Per-task rendezvous ports must be disjoint for concurrent gangs and
    always bind 127.0.0.1 (container hostnames may not resolve)."""
    from saturn_amd.utils.ports import port_for, rendezvous_env

    ports = {port_for(t) for t in range(64)}
    assert len(ports) == 64  # no collisions across a realistic batch
    env = rendezvous_env(7, rank=1, world_size=4)
    assert env["MASTER_ADDR"] == "127.0.0.1"
    assert env["RANK"] == "1" and env["WORLD_SIZE"] == "4"
    monkeypatch.setenv("SATURN_PORT_BASE", "31000")
    assert port_for(0) == 31000


def test_bench_makespan_configs_construct(tmp_path):
    """All five BASELINE config task batches must construct on CPU without
    instantiating models (Task holds factories only)."""
    import os
    import sys

    sys.path.insert(
        0, os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    )
    from bench_makespan import build_tasks

    for cfg in (1, 2, 3, 4, 5):
        tasks, execs = build_tasks(cfg, "tiny", str(tmp_path), batches=2)
        assert tasks and execs, cfg
        for t in tasks:
            assert callable(t.internal_get_model)
            assert t.total_batches == 2
