from .backbones import SSCDModel, ResNet50, VGG16, load_sscd
from .dino import VisionTransformer, vit_small, vit_base, load_dino
from .clip_model import CLIPModel, load_clip, gen_clipscore
from .features import extract_features
from .similarity import (l2_normalize, sim_matrix, einsum_in_chunks, topk_stats,
                         top_matches, similarity_histogram, tv_loss,
                         glcm_entropy, jpeg_size, pearson, pearson_with_p)

__all__ = [
    "SSCDModel", "ResNet50", "VGG16", "load_sscd",
    "VisionTransformer", "vit_small", "vit_base", "load_dino",
    "CLIPModel", "load_clip", "gen_clipscore",
    "extract_features",
    "l2_normalize", "sim_matrix", "einsum_in_chunks", "topk_stats",
    "top_matches", "similarity_histogram", "tv_loss", "glcm_entropy",
    "jpeg_size", "pearson", "pearson_with_p",
]
