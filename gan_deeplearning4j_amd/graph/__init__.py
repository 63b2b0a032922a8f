from .layers import (
    ActivationLayer,
    BatchNormLayer,
    Conv2dLayer,
    ConvTranspose2dLayer,
    DenseLayer,
    MaxPool2dLayer,
    MergeVertex,
    OutputLayer,
    Upsampling2dLayer,
    CnnToFeedForwardPreProcessor,
    FeedForwardToCnnPreProcessor,
    ReshapeVertex,
)
from .builder import ComputationGraph, GraphBuilder, InputType
from .transfer import FineTuneConfiguration, TransferLearningBuilder
from .serialization import ModelSerializer

__all__ = [
    "GraphBuilder",
    "ComputationGraph",
    "InputType",
    "DenseLayer",
    "Conv2dLayer",
    "ConvTranspose2dLayer",
    "BatchNormLayer",
    "MaxPool2dLayer",
    "MergeVertex",
    "Upsampling2dLayer",
    "ActivationLayer",
    "OutputLayer",
    "FeedForwardToCnnPreProcessor",
    "CnnToFeedForwardPreProcessor",
    "ReshapeVertex",
    "TransferLearningBuilder",
    "FineTuneConfiguration",
    "ModelSerializer",
]
