"""Vertex-range sharding arithmetic for multi-GPU PageRank.

Rows (destination vertices) are split into `world` equal contiguous ranges,
padded so every rank's slice has the same length (ncclAllGather needs equal
counts); the padded tail holds no real rows. Used by bench.py and by the
C library's dist path, and covered by the gloo CPU tests.
"""


def shard_size(n_vertices, world):
    return (n_vertices + world - 1) // world


def shard_range(n_vertices, world, rank):
    """Padded range [begin, end) for `rank`; end may exceed n_vertices."""
    s = shard_size(n_vertices, world)
    return rank * s, (rank + 1) * s


def shard_range_clamped(n_vertices, world, rank):
    b, e = shard_range(n_vertices, world, rank)
    return min(b, n_vertices), min(e, n_vertices)
