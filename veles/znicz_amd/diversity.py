"""Kernel-diversity analysis.

Parity target: reference ``znicz/diversity.py``
(``get_similar_kernels``): detect near-duplicate convolution kernels by
normalized cross-correlation, so redundant filters can be reported or
re-initialized.
"""

import numpy


def normalize_kernels(weights):
    """[K, ...] -> [K, D] rows with zero mean / unit norm."""
    w = numpy.asarray(weights, numpy.float64)
    flat = w.reshape(w.shape[0], -1)
    flat = flat - flat.mean(axis=1, keepdims=True)
    norm = numpy.linalg.norm(flat, axis=1, keepdims=True)
    norm[norm == 0] = 1.0
    return flat / norm


def similarity_matrix(weights):
    f = normalize_kernels(weights)
    return f.dot(f.T)


def get_similar_kernels(weights, threshold=0.75):
    """Groups of kernel indices whose pairwise |corr| > threshold
    (reference API)."""
    sim = numpy.abs(similarity_matrix(weights))
    k = sim.shape[0]
    visited = numpy.zeros(k, bool)
    groups = []
    for i in range(k):
        if visited[i]:
            continue
        near = [j for j in range(k)
                if j != i and not visited[j] and sim[i, j] > threshold]
        if near:
            group = [i] + near
            for j in group:
                visited[j] = True
            groups.append(group)
    return groups
