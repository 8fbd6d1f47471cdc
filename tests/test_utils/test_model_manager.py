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
