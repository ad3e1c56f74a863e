"""Optical-flow pre/post-processing: overlapping patch grid (>= min-overlap),
per-pixel 3x3-neighborhood feature stacking -> (patches, 2, 27, H, W), and
distance-weighted recombination of per-patch flow predictions (x flow_scale_factor).
Parity: reference data/vision/optical_flow.py:16-253 (the deepmind colab tiling).
cv2 is only needed for HSV rendering and video output (lazy import).
"""
from __future__ import annotations

import itertools
import math
from pathlib import Path
from typing import List, Tuple, Union

import numpy as np
import torch
import torch.nn.functional as F

from perceiver_amd.data.vision.video_utils import write_video


class OpticalFlowProcessor:
    def __init__(self, patch_size: Tuple[int, int], patch_min_overlap: int = 20,
                 flow_scale_factor: int = 20):
        if patch_min_overlap >= patch_size[0] or patch_min_overlap >= patch_size[1]:
            raise ValueError(
                f"Overlap should be smaller than the patch size "
                f"(patch-size='{patch_size}', patch_min_overlap='{patch_min_overlap}')."
            )
        self.patch_size = tuple(patch_size)
        self.patch_min_overlap = patch_min_overlap
        self.flow_scale_factor = flow_scale_factor

    # ------------------------------------------------------------ preprocessing
    @staticmethod
    def _to_tensor(x) -> torch.Tensor:
        if isinstance(x, torch.Tensor):
            return x
        if isinstance(x, np.ndarray):
            return torch.from_numpy(x)
        raise ValueError("Invalid input type. Provide input as np.array or torch.Tensor.")

    @staticmethod
    def _normalize(img: torch.Tensor) -> torch.Tensor:
        return img / 255.0 * 2 - 1

    def _transform(self, img: torch.Tensor) -> torch.Tensor:
        x = self._normalize(img).to(torch.float32)
        if x.shape[-1] == 3:  # channels-last -> channels-first
            x = x.permute(2, 0, 1).contiguous()
        return x

    @staticmethod
    def _pad_same(x: torch.Tensor, kernel: int, stride: int = 1, dilation: int = 1) -> torch.Tensor:
        *_, h, w = x.shape
        h2, w2 = math.ceil(h / stride), math.ceil(w / stride)
        pad_row = (h2 - 1) * stride + (kernel - 1) * dilation + 1 - h
        pad_col = (w2 - 1) * stride + (kernel - 1) * dilation + 1 - w
        return F.pad(x, (pad_col // 2, pad_col - pad_col // 2, pad_row // 2, pad_row - pad_row // 2))

    def _extract_image_patches(self, x: torch.Tensor, kernel: int, stride: int = 1,
                               dilation: int = 1) -> torch.Tensor:
        """tf.image.extract_patches with SAME padding: stacks each pixel's kernel x
        kernel neighborhood into the channel dim."""
        b = x.shape[0]
        x = self._pad_same(x, kernel, stride, dilation)
        patches = x.unfold(2, kernel, stride).unfold(3, kernel, stride)
        patches = patches.permute(0, 4, 5, 1, 2, 3).contiguous()
        return patches.view(b, -1, patches.shape[-2], patches.shape[-1])

    def _compute_patch_grid_indices(self, img_shape: Tuple[int, ...]) -> List[Tuple[int, int]]:
        ys = list(range(0, img_shape[0], self.patch_size[0] - self.patch_min_overlap))
        xs = list(range(0, img_shape[1], self.patch_size[1] - self.patch_min_overlap))
        ys[-1] = img_shape[0] - self.patch_size[0]
        xs[-1] = img_shape[1] - self.patch_size[1]
        return list(itertools.product(ys, xs))

    def _preprocess(self, image_pair, grid_indices) -> torch.Tensor:
        img1 = self._to_tensor(image_pair[0])
        img2 = self._to_tensor(image_pair[1])
        if img1.shape != img2.shape:
            raise ValueError(
                f"Shapes of images must match. (shape image1='{img1.shape}', shape image2='{img2.shape}')"
            )
        height, width = img1.shape[0], img1.shape[1]
        if height < self.patch_size[0]:
            raise ValueError(f"Height of image (height='{height}') must be at least {self.patch_size[0]}.")
        if width < self.patch_size[1]:
            raise ValueError(f"Width of image (width='{width}') must be at least {self.patch_size[1]}.")

        pair = torch.stack([self._transform(img1), self._transform(img2)], dim=0)
        patch_features = []
        for y, x in grid_indices:
            patch = pair[..., y: y + self.patch_size[0], x: x + self.patch_size[1]]
            patch_features.append(self._extract_image_patches(patch, kernel=3).float())
        return torch.stack(patch_features, dim=0)

    def preprocess(self, image_pair) -> torch.Tensor:
        """-> (nr_patches, 2, 27, patch_h, patch_w)"""
        grid_indices = self._compute_patch_grid_indices(image_pair[0].shape)
        return self._preprocess(image_pair, grid_indices)

    def preprocess_batch(self, image_pairs) -> torch.Tensor:
        """-> (batch, nr_patches, 2, 27, patch_h, patch_w)"""
        grid_indices = self._compute_patch_grid_indices(image_pairs[0][0].shape)
        return self._preprocess_batch(image_pairs, grid_indices)

    def _preprocess_batch(self, image_pairs, grid_indices) -> torch.Tensor:
        shapes = []
        for image1, image2 in image_pairs:
            shapes += [tuple(image1.shape), tuple(image2.shape)]
        if len(set(shapes)) > 1:
            raise ValueError("Shapes of images must match. Not all input images have the same shape.")
        return torch.stack([self._preprocess(p, grid_indices) for p in image_pairs], dim=0)

    # ------------------------------------------------------------ postprocessing
    def postprocess(self, predictions: torch.Tensor, img_shape: Tuple[int, ...]) -> torch.Tensor:
        """Distance-weighted blending of per-patch predictions -> (batch, H, W, 2)."""
        height, width = img_shape[0], img_shape[1]
        grid_indices = self._compute_patch_grid_indices(img_shape)
        prediction_batch = predictions.unsqueeze(0).cpu() if predictions.dim() == 4 else predictions.cpu()

        b, p, *_ = prediction_batch.shape
        if p != len(grid_indices):
            raise ValueError(
                f"Number of patches in the input does not match the number of calculated patches based "
                f"on the supplied image size (nr_patches='{p}', calculated={len(grid_indices)})."
            )

        flow_batch = []
        for prediction in prediction_batch:
            flow = torch.zeros(1, height, width, 2, dtype=torch.float32)
            flow_weights = torch.zeros(1, height, width, 1, dtype=torch.float32)
            for flow_patch, (y, x) in zip(prediction, grid_indices):
                flow_patch = flow_patch * self.flow_scale_factor
                wy, wx = torch.meshgrid(torch.arange(self.patch_size[0]),
                                        torch.arange(self.patch_size[1]), indexing="ij")
                wx = torch.minimum(wx + 1, self.patch_size[1] - wx)
                wy = torch.minimum(wy + 1, self.patch_size[0] - wy)
                weights = torch.minimum(wx, wy).view(1, self.patch_size[0], self.patch_size[1], 1)

                pad = (0, 0, x, width - x - self.patch_size[1], y, height - y - self.patch_size[0], 0, 0)
                flow += F.pad(flow_patch * weights, pad, "constant", 0)
                flow_weights += F.pad(weights, pad, "constant", 0)
            flow /= flow_weights
            flow_batch.append(flow)
        return torch.concat(flow_batch, dim=0)

    def process(self, model, image_pairs, batch_size: int) -> torch.Tensor:
        """preprocess -> micro-batched model inference -> postprocess."""
        image_shape = image_pairs[0][0].shape
        grid_indices = self._compute_patch_grid_indices(image_shape)

        predictions = []
        with torch.no_grad():
            for i in range(0, len(image_pairs), batch_size):
                feats = self._preprocess_batch(image_pairs[i: i + batch_size], grid_indices)
                feats = feats.flatten(0, 1)  # (b p) t c h w
                for j in range(0, feats.shape[0], batch_size):
                    pred = model(feats[j: j + batch_size])
                    predictions.append(pred.cpu().detach())

        flow_predictions = torch.concat(predictions, dim=0)
        flow_predictions = flow_predictions.unflatten(0, (len(image_pairs), -1))
        return self.postprocess(flow_predictions, image_shape)


def render_optical_flow(flow: np.ndarray) -> np.ndarray:
    """HSV visualization of a flow field (hue = angle, saturation = magnitude)."""
    import cv2

    hsv = np.zeros((flow.shape[0], flow.shape[1], 3), dtype=np.uint8)
    mag, ang = cv2.cartToPolar(flow[..., 0], flow[..., 1])
    hsv[..., 0] = ang / np.pi / 2 * 180
    hsv[..., 1] = np.clip(mag * 255 / 24, 0, 255)
    hsv[..., 2] = 255
    return cv2.cvtColor(hsv, cv2.COLOR_HSV2RGB)


def write_optical_flow_video(video_path: Path, frames: List[torch.Tensor], fps: int = 30) -> None:
    write_video(video_path=video_path, frames=[render_optical_flow(f.numpy()) for f in frames], fps=fps)
