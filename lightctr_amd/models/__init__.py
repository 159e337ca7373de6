from .fm import FMModel, FMTrainer, FMHyper
from .ffm import FFMModel, FFMTrainer, FFMHyper
from .nfm import NFMModel, NFMTrainer, NFMHyper
from .wide_deep import WideDeepModel, WideDeepTrainer, WideDeepHyper
from .mlp import MLP, DenseLayer
from .gbm import GBMModel, GBMHyper
from .gmm import GMMModel, GMMHyper
from .plsa import PLSAModel, PLSAHyper
from .embedding import EmbedModel, EmbedHyper, vocab_from_tokens
from .vae import VAEModel, VAEHyper
from .cnn import CNNModel, CNNHyper, Conv2DLayer, MaxPool2DLayer
from .rnn import RNNModel, RNNHyper, LSTMUnit, AttentionUnit
from .lr import LRModel, LRHyper, LRTrainer

__all__ = [
    "FMModel", "FMTrainer", "FMHyper",
    "FFMModel", "FFMTrainer", "FFMHyper",
    "NFMModel", "NFMTrainer", "NFMHyper",
    "WideDeepModel", "WideDeepTrainer", "WideDeepHyper",
    "MLP", "DenseLayer",
    "GBMModel", "GBMHyper",
    "GMMModel", "GMMHyper",
    "PLSAModel", "PLSAHyper",
    "EmbedModel", "EmbedHyper", "vocab_from_tokens",
    "VAEModel", "VAEHyper",
    "CNNModel", "CNNHyper", "Conv2DLayer", "MaxPool2DLayer",
    "RNNModel", "RNNHyper", "LSTMUnit", "AttentionUnit",
    "LRModel", "LRHyper", "LRTrainer",
]
