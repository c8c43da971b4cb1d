from greptimedb_amd.vector.ivf import build_ivf, ivf_candidates, kmeans

__all__ = ["build_ivf", "ivf_candidates", "kmeans"]
