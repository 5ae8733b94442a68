"""torch.hub entry point — API-compatible with the reference hubconf.py
(/root/reference/hubconf.py:37-96): `waternet(pretrained=True, device=None)`
returns the tuple (preprocess, postprocess, model). The tuple ORDER is the
contract, and `pretrained` defaults True as in the reference (hubconf.py:37).

Pretrained weights come from the reference's published checkpoint URL with
torch.hub's hash check (the filename embeds the hash `daa0ee`, so
check_hash=True validates the download — reference hubconf.py:5,78-83).
In an offline environment the download raises a clear error; pass
`checkpoint=<local .pt path>` instead, or `pretrained=False` for random
init.
"""

dependencies = ["torch", "numpy"]

# The reference's published checkpoint (hubconf.py:5). The filename's
# -daa0ee suffix is the sha256 prefix torch.hub verifies with check_hash.
WEIGHTS_URL = (
    "https://www.dropbox.com/s/j8ida1d86hy5tm4/"
    "waternet_exported_state_dict-daa0ee.pt?dl=1"
)
WEIGHTS_FILE = "waternet_exported_state_dict-daa0ee.pt"


def _load_pretrained(model, device):
    import torch

    try:
        sd = torch.hub.load_state_dict_from_url(
            WEIGHTS_URL,
            map_location=device if device is not None else "cpu",
            file_name=WEIGHTS_FILE,
            check_hash=True,
        )
    except Exception as e:  # noqa: BLE001 — no network in some deployments
        raise RuntimeError(
            "Could not download the pretrained WaterNet checkpoint "
            f"({WEIGHTS_URL}): {e!r}. If this environment has no network "
            "access, pass checkpoint=<path to a WaterNet state_dict .pt> "
            "or pretrained=False."
        ) from e
    model.load_state_dict(sd)


def waternet(pretrained=True, device=None, checkpoint=None):
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
        _load_pretrained(model, device)

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
