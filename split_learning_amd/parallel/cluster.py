"""Client clustering by label distribution (reference src/Cluster.py:5-21):
L1-normalised label-count vectors -> KMeans(num_cluster, random_state=42)."""

from __future__ import annotations

import numpy as np


def clustering_algorithm(label_counts, num_cluster: int, algorithm: str = "KMeans"):
    return k_means(label_counts, num_cluster)


def k_means(label_counts, num_cluster: int):
    from sklearn.cluster import KMeans
    from sklearn.preprocessing import normalize

    x = normalize(np.asarray(label_counts, dtype=float), norm="l1", axis=1)
    km = KMeans(n_clusters=num_cluster, random_state=42)
    km.fit(x)
    labels = km.labels_
    infor_cluster = [[int(c)] for c in np.bincount(labels, minlength=num_cluster)]
    return labels, infor_cluster
