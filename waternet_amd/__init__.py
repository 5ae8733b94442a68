"""waternet_amd — MI355X-native WaterNet underwater-image-enhancement engine.

Public API (mirrors the reference's import surface plus the engine layers):

    from waternet_amd import WaterNet, PerceptualModel, transform
    from waternet_amd.engine.trainer import train_one_epoch, eval_one_epoch
    from waternet_amd.engine.inferencer import InferenceEngine
"""

from waternet_amd.data.transforms import transform  # noqa: F401
from waternet_amd.models.vgg import PerceptualModel  # noqa: F401
from waternet_amd.models.waternet import WaterNet  # noqa: F401

__all__ = ["WaterNet", "PerceptualModel", "transform"]
