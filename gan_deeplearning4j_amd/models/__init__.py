from .reference_protocol import (
    build_discriminator,
    build_frozen_generator,
    build_stacked_gan,
    build_transfer_classifier,
    GAN_TO_GEN_SYNC,
    DIS_TO_GAN_SYNC,
    DIS_TO_CV_SYNC,
)
from .cgan import build_cgan
from .dcgan import build_dcgan
from .mlp_gan import build_mlp_gan

__all__ = [
    "build_discriminator",
    "build_frozen_generator",
    "build_stacked_gan",
    "build_transfer_classifier",
    "build_cgan",
    "build_dcgan",
    "build_mlp_gan",
    "GAN_TO_GEN_SYNC",
    "DIS_TO_GAN_SYNC",
    "DIS_TO_CV_SYNC",
]
