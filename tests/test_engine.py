"""Driver config 1: one train step on CPU, losses finite (plumbing)."""
import torch

from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
from mine_amd.engine import SynthesisTask
from mine_amd.engine.checkpoint import restore_model, save_checkpoint


def _items(cfg, n=2):
    ds = SyntheticMPIDataset(cfg, length=n)
    return collate_src_tgt([ds[i] for i in range(n)])


def test_one_train_step_cpu(tiny_config):
    task = SynthesisTask(tiny_config, device="cpu")
    loss_dict = task.train_step(_items(tiny_config))
    for k, v in loss_dict.items():
        v = float(v)
        assert v == v, f"{k} is NaN"
        assert abs(v) < 1e6, f"{k} diverged: {v}"
    assert float(loss_dict["loss"]) > 0


def test_two_steps_update_params(tiny_config):
    task = SynthesisTask(tiny_config, device="cpu")
    p0 = task.decoder.dispconvs["0"].conv.weight.detach().clone()
    task.train_step(_items(tiny_config))
    p1 = task.decoder.dispconvs["0"].conv.weight.detach().clone()
    assert (p0 - p1).abs().max() > 0


def test_grads_flow_to_all_params(tiny_config):
    task = SynthesisTask(tiny_config, device="cpu")
    task.set_data(_items(tiny_config))
    loss_dict, _ = task.loss_fcn(is_val=False)
    loss_dict["loss"].backward()
    for name, p in list(task.backbone.named_parameters()) + \
            list(task.decoder.named_parameters()):
        assert p.grad is not None, name
    # the rendering path must feed gradient into sigma (dispconv weights)
    g = task.decoder.dispconvs["0"].conv.weight.grad
    assert g.abs().max() > 0


def test_eval_runs(tiny_config):
    from torch.utils.data import DataLoader
    task = SynthesisTask(tiny_config, device="cpu")
    ds = SyntheticMPIDataset(tiny_config, is_validation=True, length=2)
    dl = DataLoader(ds, batch_size=2, collate_fn=collate_src_tgt)
    task.run_eval(dl)
    assert task.val_losses["psnr_tgt"].count > 0


def test_checkpoint_roundtrip(tiny_config, tmp_path):
    task = SynthesisTask(tiny_config, device="cpu")
    path = str(tmp_path / "checkpoint.pth")
    save_checkpoint(path, task.backbone, task.decoder, task.optimizer)
    state = torch.load(path, weights_only=False)
    assert set(state.keys()) == {"backbone", "decoder", "optimizer"}

    task2 = SynthesisTask(tiny_config, device="cpu")
    restore_model(path, task2.backbone, task2.decoder, task2.optimizer)
    for (k1, v1), (k2, v2) in zip(task.backbone.state_dict().items(),
                                  task2.backbone.state_dict().items()):
        assert k1 == k2
        torch.testing.assert_close(v1, v2)


def test_checkpoint_module_prefix_tolerated(tiny_config, tmp_path):
    """Reference checkpoints carry 'module.' prefixes from DDP wrapping
    (ref utils.py:53-54) — restore must strip them."""
    task = SynthesisTask(tiny_config, device="cpu")
    path = str(tmp_path / "checkpoint.pth")
    state = {
        "backbone": {"module." + k: v for k, v in task.backbone.state_dict().items()},
        "decoder": {"module." + k: v for k, v in task.decoder.state_dict().items()},
    }
    torch.save(state, path)
    task2 = SynthesisTask(tiny_config, device="cpu")
    restore_model(path, task2.backbone, task2.decoder)
    torch.testing.assert_close(task2.backbone.conv1.weight, task.backbone.conv1.weight)


def test_fixed_disparity_mode(tiny_config):
    from mine_amd.engine.task import get_disparity_list
    cfg = tiny_config.replace(**{"mpi.fix_disparity": True})
    d = get_disparity_list(cfg, 3, torch.device("cpu"))
    assert d.shape == (3, cfg["mpi.num_bins_coarse"])
    torch.testing.assert_close(d[0], d[1])
    assert (d[:, :-1] > d[:, 1:]).all()


def test_coarse_to_fine_path(tiny_config):
    cfg = tiny_config.replace(**{"mpi.num_bins_fine": 4})
    task = SynthesisTask(cfg, device="cpu")
    task.set_data(_items(cfg))
    endpoints = task.network_forward()
    S = cfg["mpi.num_bins_coarse"] + 4
    assert endpoints["disparity_all_src"].shape[1] == S
    d = endpoints["disparity_all_src"]
    assert (d[:, :-1] >= d[:, 1:]).all()  # sorted descending
    assert endpoints["mpi_all_src_list"][0].shape[1] == S


def test_nan_guard_skips_update():
    import torch
    from mine_amd.config import default_config
    from mine_amd.data import SyntheticMPIDataset, collate_src_tgt
    from mine_amd.engine import SynthesisTask

    cfg = default_config(**{
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 8, "training.amp_dtype": "fp32",
    })
    ds = SyntheticMPIDataset(cfg, length=2)
    items = collate_src_tgt([ds[0], ds[1]])
    task = SynthesisTask(cfg, device="cpu")

    # poison the input -> non-finite loss
    bad = (dict(items[0]), items[1])
    bad[0]["img"] = items[0]["img"] * float("nan")
    w0 = task.decoder.dispconvs["0"].conv.weight.detach().clone()
    loss = task.train_step((bad[0], bad[1]))
    assert not torch.isfinite(loss["loss"])
    # device-side guard: gradients are zeroed before the optimizer step,
    # so no NaN ever enters the parameters or the Adam moments (the step
    # itself still runs — momentum-decay drift only, no host sync)
    for p in list(task.backbone.parameters()) + list(task.decoder.parameters()):
        assert torch.isfinite(p).all()
        assert p.grad is None or torch.isfinite(p.grad).all()
    assert int(task._nan_skip_count.item()) == 1

    # a healthy step still updates and parameters stay finite
    loss2 = task.train_step(items)
    assert torch.isfinite(loss2["loss"])
    for p in task.decoder.parameters():
        assert torch.isfinite(p).all()
    assert int(task._nan_skip_count.item()) == 1
    assert not torch.equal(task.decoder.dispconvs["0"].conv.weight, w0)
