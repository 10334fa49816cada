from .embed import generate_embeddings, extract_features_custom, SyntheticLAIONDataset
from .search import stream_top1, dump_matches, sharded_topk, distributed_knn

__all__ = ["generate_embeddings", "extract_features_custom", "SyntheticLAIONDataset",
           "stream_top1", "dump_matches", "sharded_topk", "distributed_knn"]
