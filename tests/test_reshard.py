"""World-size-independent optimizer restore: canonical interchange between
FusedAdamW ("groups", replicated) and ShardedEngine ("shards", world-sliced),
including the checkpointer's cross-world load path (2-rank save -> 1-process
resume and the reverse)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _init(rank, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)


def _run(fn, port, *args):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(r, port) + args) for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    for p in procs:
        assert p.exitcode == 0, f"child exited with {p.exitcode}"


def _make_model():
    torch.manual_seed(7)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.GELU(), torch.nn.Linear(32, 16),
        torch.nn.LayerNorm(16),
    )


def _make_groups(model):
    decay, no_decay = [], []
    for name, p in model.named_parameters():
        (no_decay if (name.endswith("bias") or "3." in name) else decay).append(p)
    return [
        {"params": decay, "names": [f"d{i}" for i in range(len(decay))],
         "submodel": "backbone", "lr_multiplier": 1.0, "wd_multiplier": 1.0,
         "is_last_layer": False},
        {"params": no_decay, "names": [f"n{i}" for i in range(len(no_decay))],
         "submodel": "backbone", "lr_multiplier": 0.5, "wd_multiplier": 0.0,
         "is_last_layer": False},
    ]


def _replicated_run(n_steps=2):
    """Single-process FusedAdamW on the rank-averaged gradients."""
    from dinov3_amd.train.optim import FusedAdamW

    model = _make_model()
    opt = FusedAdamW(_make_groups(model), use_master_weights=False)
    for step in range(n_steps):
        losses = []
        for r in range(WORLD):
            torch.manual_seed(100 * step + r)
            x = torch.randn(8, 16)
            losses.append((model(x) ** 2).mean())
        (sum(losses) / WORLD).backward()
        opt.step(lr=0.05, weight_decay=0.1)
        opt.zero_grad()
    return model, opt


def _sharded_save_worker(rank, port, tmpdir):
    _init(rank, port)
    from dinov3_amd.parallel.fsdp import ShardedEngine

    model = _make_model()
    engine = ShardedEngine(_make_groups(model), align=4)
    for step in range(2):
        torch.manual_seed(100 * step + rank)
        x = torch.randn(8, 16)
        (model(x) ** 2).mean().backward()
        engine.finalize_backward()
        engine.step(lr=0.05, weight_decay=0.1)
        engine.zero_grad()
    torch.save({"iteration": 1, "model": model.state_dict(),
                "optimizer": engine.state_dict(), "world_size": WORLD},
               os.path.join(tmpdir, f"rank_{rank}.pth"))
    dist.destroy_process_group()


def _canon_allclose(a, b, atol=2e-6):
    assert a["step_count"] == b["step_count"]
    assert set(a["groups"]) == set(b["groups"])
    for key in a["groups"]:
        for field in ("exp_avg", "exp_avg_sq", "master"):
            va, vb = a["groups"][key][field], b["groups"][key][field]
            if va is None or vb is None:
                assert va is None and vb is None, (key, field)
                continue
            assert torch.allclose(va, vb, atol=atol), (
                key, field, (va - vb).abs().max().item())


def test_sharded_canonical_matches_replicated(tmp_path):
    """canonical(2-rank shard states) == canonical(replicated reference)."""
    from dinov3_amd.train.optim_state import to_canonical

    _run(_sharded_save_worker, 29661, str(tmp_path))
    states = [torch.load(tmp_path / f"rank_{r}.pth", weights_only=False)["optimizer"]
              for r in range(WORLD)]
    canon_sharded = to_canonical(states)
    _, ref_opt = _replicated_run()
    canon_ref = to_canonical([ref_opt.state_dict()])
    _canon_allclose(canon_sharded, canon_ref)


def test_world2_checkpoint_restores_into_single_process(tmp_path):
    """A 2-rank ShardedEngine checkpoint resumes into a world-1 FusedAdamW via
    the checkpointer path (load_optimizer_state reads every rank file)."""
    from dinov3_amd.train.optim import FusedAdamW
    from dinov3_amd.train.optim_state import load_optimizer_state, to_canonical

    _run(_sharded_save_worker, 29663, str(tmp_path))
    payload = torch.load(tmp_path / "rank_0.pth", weights_only=False)
    model = _make_model()
    model.load_state_dict(payload["model"])
    opt = FusedAdamW(_make_groups(model), use_master_weights=False)
    load_optimizer_state(opt, tmp_path, payload, rank=0, world=1)
    assert opt.step_count == 2
    _, ref_opt = _replicated_run()
    _canon_allclose(to_canonical([opt.state_dict()]),
                    to_canonical([ref_opt.state_dict()]))


def _sharded_load_worker(rank, port, tmpdir):
    _init(rank, port)
    from dinov3_amd.parallel.fsdp import ShardedEngine
    from dinov3_amd.train.optim_state import load_optimizer_state, to_canonical

    payload = torch.load(os.path.join(tmpdir, "groups_ckpt.pth"), weights_only=False)
    model = _make_model()
    model.load_state_dict(payload["model"])
    engine = ShardedEngine(_make_groups(model), align=4)
    load_optimizer_state(engine, tmpdir, payload, rank=rank, world=WORLD)
    assert engine.step_count == 2
    # the engine's shards, re-canonicalized across ranks, must equal the source
    states = [None] * WORLD
    obj = [engine.state_dict()]
    gathered = [[None] for _ in range(WORLD)]
    dist.all_gather_object(gathered, obj[0])
    states = gathered
    canon = to_canonical(states)
    src = to_canonical([payload["optimizer"]])
    _canon_allclose({"step_count": canon["step_count"],
                     "groups": {k: {f: (v[f] if f != "master" else None)
                                    for f in ("exp_avg", "exp_avg_sq", "master")}
                                for k, v in canon["groups"].items()}},
                    {"step_count": src["step_count"],
                     "groups": {k: {f: (v[f] if f != "master" else None)
                                    for f in ("exp_avg", "exp_avg_sq", "master")}
                                for k, v in src["groups"].items()}})
    dist.destroy_process_group()


def test_world1_checkpoint_restores_into_world2(tmp_path):
    """A single-process FusedAdamW checkpoint loads into a 2-rank
    ShardedEngine; reassembling the shards reproduces the source state."""
    model, opt = _replicated_run()
    torch.save({"iteration": 1, "model": model.state_dict(),
                "optimizer": opt.state_dict(), "world_size": 1},
               tmp_path / "groups_ckpt.pth")
    _run(_sharded_load_worker, 29665, str(tmp_path))


def test_same_world_fast_path_unchanged(tmp_path):
    """Same engine + same world goes through plain load_state_dict."""
    from dinov3_amd.train.optim import FusedAdamW
    from dinov3_amd.train.optim_state import load_optimizer_state

    model, opt = _replicated_run()
    payload = {"optimizer": opt.state_dict(), "world_size": 1}
    model2 = _make_model()
    opt2 = FusedAdamW(_make_groups(model2), use_master_weights=False)
    load_optimizer_state(opt2, tmp_path, payload, rank=0, world=1)
    for g1, g2 in zip(opt.groups, opt2.groups):
        for a, b in zip(g1["exp_avg"], g2["exp_avg"]):
            assert torch.equal(a, b)


def _sharded4_load_worker(rank, port, tmpdir):
    """4-rank engine resumes the 2-rank checkpoint written by
    _sharded_save_worker (sharded -> sharded across world sizes)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=4)
    from dinov3_amd.parallel.fsdp import ShardedEngine
    from dinov3_amd.train.optim_state import load_optimizer_state, to_canonical

    payload = torch.load(os.path.join(tmpdir, "rank_0.pth"), weights_only=False)
    model = _make_model()
    model.load_state_dict(payload["model"])
    engine = ShardedEngine(_make_groups(model), align=4)
    load_optimizer_state(engine, tmpdir, payload, rank=rank, world=4)
    assert engine.step_count == 2
    gathered = [None] * 4
    dist.all_gather_object(gathered, engine.state_dict())
    canon = to_canonical(gathered)
    src = to_canonical([torch.load(os.path.join(tmpdir, f"rank_{r}.pth"),
                                   weights_only=False)["optimizer"] for r in range(2)])
    for key in src["groups"]:
        for field in ("exp_avg", "exp_avg_sq"):
            a, b = canon["groups"][key][field], src["groups"][key][field]
            assert torch.allclose(a, b, atol=2e-6), (key, field)
    dist.destroy_process_group()


def test_world2_checkpoint_restores_into_world4(tmp_path):
    _run(_sharded_save_worker, 29667, str(tmp_path))
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_sharded4_load_worker, args=(r, 29669, str(tmp_path)))
             for r in range(4)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
    for p in procs:
        assert p.exitcode == 0, f"child exited with {p.exitcode}"
