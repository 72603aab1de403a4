"""R9: MatConvNet imagenet-vgg-f.mat import.

No network access, so this builds a synthetic .mat with MatConvNet's
net.layers structure (struct array with name/type/weights cells) and checks
the weights land in the right VGG-F parameters.
"""

import numpy as np
import torch
from scipy.io import savemat

from distributed_vgg_f_amd.config import Config
from distributed_vgg_f_amd.engine.pretrained import load_matconvnet_vggf
from distributed_vgg_f_amd.models import build_model

SHAPES = {
    "conv1": (11, 11, 3, 64), "conv2": (5, 5, 64, 256),
    "conv3": (3, 3, 256, 256), "conv4": (3, 3, 256, 256),
    "conv5": (3, 3, 256, 256),
    "fc6": (6, 6, 256, 4096), "fc7": (1, 1, 4096, 4096),
    "fc8": (1, 1, 4096, 1000),
}


def _synthetic_mat(path):
    rng = np.random.default_rng(0)
    layers = []
    for name, shape in SHAPES.items():
        f = rng.standard_normal(shape).astype(np.float32) * 0.01
        b = rng.standard_normal(shape[-1]).astype(np.float32)
        weights = np.empty((1, 2), dtype=object)
        weights[0, 0] = f
        weights[0, 1] = b
        layers.append({"name": name, "type": "conv", "weights": weights})
        if name.startswith("conv"):
            layers.append({"name": name.replace("conv", "relu"),
                           "type": "relu"})
    arr = np.empty((1, len(layers)), dtype=object)
    for i, l in enumerate(layers):
        arr[0, i] = l
    savemat(path, {"layers": arr})
    return {n: (f, None) for n, f in SHAPES.items()}


def test_matconvnet_import(tmp_path):
    path = str(tmp_path / "imagenet-vgg-f.mat")
    _synthetic_mat(path)
    cfg = Config(model="vggf")
    torch.manual_seed(0)
    model = build_model(cfg)
    n = load_matconvnet_vggf(path, model)
    assert n == 16, n  # 8 layers x (weight, bias)
    # spot-check a value and a shape mapping
    from scipy.io import loadmat
    mat = loadmat(path)
    params = dict(model.named_parameters())
    # conv2 filters landed in features.3.weight (HWIO straight copy)
    cell = None
    for c in np.ravel(mat["layers"]):
        e = np.ravel(c)[0]
        if str(np.ravel(e["name"])[0]) == "conv2":
            cell = np.asarray(np.ravel(e["weights"])[0], dtype=np.float32)
    assert cell is not None
    assert torch.allclose(params["features.3.weight"],
                          torch.from_numpy(cell))
    # fc6 reshaped 6x6x256x4096 -> [9216, 4096]
    assert params["fc6.weight"].shape == (9216, 4096)
