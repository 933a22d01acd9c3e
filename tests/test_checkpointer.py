import torch

from dinov3_amd.checkpointer import find_latest_checkpoint, load_checkpoint, save_checkpoint


def _tiny_model():
    torch.manual_seed(0)
    return torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))


def test_roundtrip(tmp_path):
    model = _tiny_model()
    out = str(tmp_path)
    save_checkpoint(out, 10, model, max_to_keep=3)
    latest = find_latest_checkpoint(out)
    assert latest is not None and latest.name == "10"
    model2 = _tiny_model()
    with torch.no_grad():
        for p in model2.parameters():
            p.zero_()
    payload = load_checkpoint(latest, model2)
    assert payload["iteration"] == 10
    for p1, p2 in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p1, p2)


def test_retention(tmp_path):
    model = _tiny_model()
    out = str(tmp_path)
    for it in (1, 2, 3, 4, 5):
        save_checkpoint(out, it, model, max_to_keep=2)
    import os

    kept = sorted(os.listdir(os.path.join(out, "ckpt")))
    assert kept == ["4", "5"]
    assert find_latest_checkpoint(out).name == "5"


def test_keep_every(tmp_path):
    model = _tiny_model()
    out = str(tmp_path)
    for it in (10, 20, 30):
        save_checkpoint(out, it, model, max_to_keep=1, keep_every=20)
    import os

    kept = sorted(os.listdir(os.path.join(out, "ckpt")), key=int)
    assert kept == ["20", "30"]


def test_partial_restore(tmp_path):
    model = _tiny_model()
    out = str(tmp_path)
    save_checkpoint(out, 1, model)
    bigger = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2), torch.nn.Linear(2, 2))
    payload = load_checkpoint(find_latest_checkpoint(out), bigger, strict=False)
    assert payload["iteration"] == 1


def test_optimizer_state_roundtrip(tmp_path):
    from dinov3_amd.train.optim import FusedAdamW

    model = _tiny_model()
    groups = [{"params": list(model.parameters()), "names": ["a", "b", "c", "d"],
               "submodel": "backbone", "lr_multiplier": 1.0, "wd_multiplier": 1.0,
               "is_last_layer": False}]
    opt = FusedAdamW(groups, use_master_weights=False)
    for p in model.parameters():
        p.grad = torch.randn_like(p)
    opt.step(lr=0.01, weight_decay=0.0)
    out = str(tmp_path)
    save_checkpoint(out, 5, model, opt)
    opt2 = FusedAdamW(groups, use_master_weights=False)
    load_checkpoint(find_latest_checkpoint(out), model, opt2)
    assert opt2.step_count == 1
    for g1, g2 in zip(opt.groups, opt2.groups):
        for m1, m2 in zip(g1["exp_avg"], g2["exp_avg"]):
            assert torch.equal(m1, m2)
