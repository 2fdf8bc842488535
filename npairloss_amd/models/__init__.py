from .googlenet import GoogLeNet
from .resnet import ResNet50
from .vit import ViTB16
from .embedding import EmbeddingNet, build_embedding_model

__all__ = ["GoogLeNet", "ResNet50", "ViTB16", "EmbeddingNet", "build_embedding_model"]
