from perceiver_amd.data.vision.common import ImagePreprocessor, channels_to_last, lift_transform
from perceiver_amd.data.vision.imagenet import ImageNetPreprocessor
from perceiver_amd.data.vision.mnist import MNISTDataModule, MNISTPreprocessor
from perceiver_amd.data.vision.optical_flow import OpticalFlowProcessor, render_optical_flow
