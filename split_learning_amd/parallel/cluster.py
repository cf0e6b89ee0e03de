"""Client clustering by label distribution (reference src/Cluster.py:5-21):
L1-normalised label-count vectors -> KMeans(num_cluster, random_state=42).
AffinityPropagation is also supported (the reference README.md:113 documents it
but the reference code never implemented it)."""

from __future__ import annotations

import numpy as np


def clustering_algorithm(label_counts, num_cluster: int, algorithm: str = "KMeans"):
    if algorithm == "AffinityPropagation":
        return affinity_propagation(label_counts, num_cluster)
    return k_means(label_counts, num_cluster)


def affinity_propagation(label_counts, num_cluster: int):
    """Cluster count is emergent; num_cluster is only a fallback cap used when
    AP degenerates (all-one-cluster with huge member counts downstream)."""
    from sklearn.cluster import AffinityPropagation
    from sklearn.preprocessing import normalize

    x = normalize(np.asarray(label_counts, dtype=float), norm="l1", axis=1)
    ap = AffinityPropagation(random_state=42).fit(x)
    labels = ap.labels_
    if labels.max() < 0:  # did not converge
        return k_means(label_counts, num_cluster)
    n = int(labels.max()) + 1
    infor = [[int(c)] for c in np.bincount(labels, minlength=n)]
    return labels, infor


def k_means(label_counts, num_cluster: int):
    from sklearn.cluster import KMeans
    from sklearn.preprocessing import normalize

    x = normalize(np.asarray(label_counts, dtype=float), norm="l1", axis=1)
    km = KMeans(n_clusters=num_cluster, random_state=42)
    km.fit(x)
    labels = km.labels_
    infor_cluster = [[int(c)] for c in np.bincount(labels, minlength=num_cluster)]
    return labels, infor_cluster
