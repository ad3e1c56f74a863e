"""Vision data pipeline tests: MNIST transforms and the optical-flow patch
tiling/recombination (no network; synthetic arrays)."""
import numpy as np
import torch

from perceiver_amd.data.vision.mnist import mnist_transform
from perceiver_amd.data.vision.optical_flow import OpticalFlowProcessor


def test_mnist_transform_shapes_and_range():
    from PIL import Image

    img = Image.fromarray((np.random.rand(28, 28) * 255).astype("uint8"))
    t = mnist_transform(normalize=True, channels_last=True)(img)
    assert t.shape == (28, 28, 1)
    assert -1.001 <= t.min() and t.max() <= 1.001
    t2 = mnist_transform(normalize=False, channels_last=False)(img)
    assert t2.shape == (1, 28, 28)
    assert 0 <= t2.min() and t2.max() <= 1


def test_optical_flow_patch_grid_and_features():
    proc = OpticalFlowProcessor(patch_size=(32, 48), patch_min_overlap=8)
    img1 = np.random.randint(0, 255, (60, 90, 3), dtype=np.uint8)
    img2 = np.random.randint(0, 255, (60, 90, 3), dtype=np.uint8)
    feats = proc.preprocess((img1, img2))
    grid = proc._compute_patch_grid_indices(img1.shape)
    assert feats.shape == (len(grid), 2, 27, 32, 48)
    # last grid entries snapped so patches fit
    assert all(y + 32 <= 60 and x + 48 <= 90 for y, x in grid)


def test_optical_flow_postprocess_blends_to_image():
    proc = OpticalFlowProcessor(patch_size=(32, 48), patch_min_overlap=8, flow_scale_factor=20)
    img_shape = (60, 90, 3)
    grid = proc._compute_patch_grid_indices(img_shape)
    # constant flow per patch -> blended result must be that constant * scale
    preds = torch.full((len(grid), 32, 48, 2), 0.05)
    flow = proc.postprocess(preds, img_shape)
    assert flow.shape == (1, 60, 90, 2)
    assert torch.allclose(flow, torch.full_like(flow, 1.0), atol=1e-5)


def test_optical_flow_process_end_to_end_tiny_model():
    class Dummy(torch.nn.Module):
        def forward(self, x):  # (b, 2, 27, h, w) -> (b, h, w, 2)
            b, _, _, h, w = x.shape
            return torch.zeros(b, h, w, 2)

    proc = OpticalFlowProcessor(patch_size=(32, 48), patch_min_overlap=8)
    img1 = np.random.randint(0, 255, (60, 90, 3), dtype=np.uint8)
    pairs = [(img1, img1), (img1, img1)]
    flow = proc.process(Dummy(), pairs, batch_size=2)
    assert flow.shape == (2, 60, 90, 2)
    assert torch.allclose(flow, torch.zeros_like(flow))
