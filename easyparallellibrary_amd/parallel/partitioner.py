"""Weighted list partitioning + repeated-block detection.

Capability parity: /root/reference/epl/parallel/partitioner.py
(partition_balance :44, partition_stages :124, find_repeated_blocks
:79-121 — op-type-histogram detection at scope depth; here the unit is a
module class instead of an op-type histogram).
"""

from collections import Counter


def partition_balance(weights, k):
    """Split ``weights`` (list of costs) into ``k`` contiguous chunks
    minimizing the maximum chunk cost (linear-partition DP; reference
    partition_balance is a greedy equivalent).  Returns list of k lists of
    indices."""
    n = len(weights)
    k = min(k, n)
    if k <= 1:
        return [list(range(n))]
    prefix = [0]
    for w in weights:
        prefix.append(prefix[-1] + w)

    def cost(i, j):
        return prefix[j] - prefix[i]

    INF = float("inf")
    dp = [[INF] * (k + 1) for _ in range(n + 1)]
    cut = [[0] * (k + 1) for _ in range(n + 1)]
    dp[0][0] = 0
    for j in range(1, k + 1):
        for i in range(1, n + 1):
            for t in range(j - 1, i):
                c = max(dp[t][j - 1], cost(t, i))
                if c < dp[i][j]:
                    dp[i][j] = c
                    cut[i][j] = t
    bounds = [n]
    i, j = n, k
    while j > 0:
        t = cut[i][j]
        bounds.append(t)
        i, j = t, j - 1
    bounds.reverse()
    return [list(range(bounds[i], bounds[i + 1])) for i in range(k)]


def find_repeated_blocks(items, key=lambda x: type(x).__name__,
                         min_repeat=3):
    """Find the dominant repeated unit in an ordered item list (reference:
    repeated-block detection by type histogram :79-121).  Returns the key
    that repeats most, or None."""
    counts = Counter(key(it) for it in items)
    best = None
    best_n = 0
    for kk, n in counts.items():
        if n >= min_repeat and n > best_n:
            best, best_n = kk, n
    return best
