import torch

from sheeprl_amd.utils.model_manager import ModelManager


def test_register_load_transition_delete(tmp_path):
    mm = ModelManager(tmp_path / "registry")
    sd = {"w": torch.randn(3, 3)}
    v1 = mm.register_model("mymodel", sd, description="first")
    assert v1 == 1
    v2 = mm.register_model("mymodel", {"w": torch.zeros(3, 3)})
    assert v2 == 2
    assert mm.get_latest_version("mymodel") == 2
    loaded = mm.load_model("mymodel", version=1)
    assert torch.allclose(loaded["w"], sd["w"])
    mm.transition_model("mymodel", stage="production")
    assert mm.get_model_info("mymodel")["stage"] == "production"
    assert mm.registered_models() == {"mymodel": [1, 2]}
    mm.delete_model("mymodel", version=1)
    assert mm.registered_models() == {"mymodel": [2]}
    assert (tmp_path / "registry" / "mymodel" / "changelog.md").exists()


def test_register_models_from_checkpoint(tmp_path):
    """The registration CLI path must honor each algo's declared
    MODELS_TO_REGISTER (wherever the algo defines it) and skip optimizers."""
    import torch.nn as nn

    from sheeprl_amd.utils.dotdict import DotDict
    from sheeprl_amd.utils.model_manager import register_models_from_checkpoint

    net = nn.Linear(3, 2)
    opt = torch.optim.Adam(net.parameters())
    net(torch.randn(1, 3)).sum().backward()
    opt.step()
    ckpt = tmp_path / "ckpt_10_0.ckpt"
    torch.save({"agent": net.state_dict(), "optimizer": opt.state_dict(), "iter_num": 10}, ckpt)

    cfg = DotDict({"algo": {"name": "a2c"}, "env": {"id": "dummy"}})
    versions = register_models_from_checkpoint(cfg, str(ckpt), tmp_path / "registry")
    assert versions == {"agent": 1}
    assert (tmp_path / "registry" / "a2c_dummy_agent").is_dir()

    # exploration-variant algo names resolve their entrypoint module too
    state = {
        k: nn.Linear(2, 2).state_dict()
        for k in ("world_model", "ensembles", "actor_task", "critic_task",
                  "target_critic_task", "actor_exploration")
    }
    ckpt2 = tmp_path / "ckpt_20_0.ckpt"
    torch.save(state, ckpt2)
    cfg2 = DotDict({"algo": {"name": "p2e_dv3_exploration"}, "env": {"id": "dummy"}})
    versions2 = register_models_from_checkpoint(cfg2, str(ckpt2), tmp_path / "registry")
    assert set(versions2) == set(state)
