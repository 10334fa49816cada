from .fid import calculate_fid_given_paths, calculate_frechet_distance, save_fid_stats
from .inception import InceptionV3
from .ipr import IPR, compute_pairwise_distances

__all__ = ["calculate_fid_given_paths", "calculate_frechet_distance", "save_fid_stats", "InceptionV3", "IPR", "compute_pairwise_distances"]
