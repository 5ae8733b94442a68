"""torch.hub entry point — API-compatible with the reference hubconf.py
(/root/reference/hubconf.py:37-96): `waternet(pretrained, device)` returns
the tuple (preprocess, postprocess, model). The tuple ORDER is the contract.

No network is available in this environment, so pretrained=True accepts a
local checkpoint path via the `checkpoint` argument instead of downloading;
with neither, it raises.
"""

dependencies = ["torch", "numpy"]


def waternet(pretrained=False, device=None, checkpoint=None):
    """
    Returns (preprocess, postprocess, model):
      preprocess(rgb_arr) -> (rgb_ten, wb_ten, he_ten, gc_ten)  NCHW [0,1]
      postprocess(model_out) -> NHWC uint8 array
      model: WaterNet on `device`
    """
    import torch

    from waternet_amd.data.bridge import arr2ten, ten2arr
    from waternet_amd.data.transforms import transform
    from waternet_amd.models.waternet import WaterNet

    model = WaterNet()

    if checkpoint is not None:
        with open(checkpoint, "rb") as f:
            model.load_state_dict(torch.load(f, map_location="cpu"))
    elif pretrained:
        raise RuntimeError(
            "No network access for pretrained weight download; pass "
            "checkpoint=<path to a WaterNet state_dict .pt> instead."
        )

    def preprocess(rgb_arr):
        wb, gc, he = transform(rgb_arr)
        rgb_ten = arr2ten(rgb_arr, add_batch_dim=True)
        wb_ten = arr2ten(wb, add_batch_dim=True)
        gc_ten = arr2ten(gc, add_batch_dim=True)
        he_ten = arr2ten(he, add_batch_dim=True)
        # Tuple order (rgb, wb, he, gc) matches hubconf.py:91 — he in the
        # `ce` slot.
        return rgb_ten, wb_ten, he_ten, gc_ten

    def postprocess(model_out):
        return ten2arr(model_out)

    return preprocess, postprocess, model.to(device)
