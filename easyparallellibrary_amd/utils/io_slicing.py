"""Input-file slicing across data-parallel replicas.

Capability parity: /root/reference/epl/parallel/graph_editor.py:149-215 +
fetch_slice_objects_proportion_to_local_num_replicas (:787-854) and the
io config section (epl/config.py io.slicing / unbalanced_io_slicing).

Each replica reads a disjoint slice of the input file list; when the list
does not divide evenly, the reference's proportional policy gives the
first ``remainder`` replicas one extra file (balanced), or — with
unbalanced_io_slicing — keeps strict contiguous proportional slices.
"""


def slice_files(files, num_replicas, replica_id, unbalanced=False,
                drop_last=False):
    files = list(files)
    if drop_last and num_replicas > 1 and len(files) % num_replicas:
        # reference io.drop_last_files: equal counts everywhere
        files = files[:len(files) - len(files) % num_replicas]
    n = len(files)
    if num_replicas <= 1:
        return files
    if n < num_replicas:
        raise ValueError(
            "cannot slice {} files over {} replicas".format(n, num_replicas))
    base = n // num_replicas
    rem = n % num_replicas
    if unbalanced:
        # strict contiguous proportional slices
        start = replica_id * base + min(replica_id, rem)
        size = base + (1 if replica_id < rem else 0)
        return files[start:start + size]
    # balanced: first `rem` replicas get one extra file (reference policy)
    start = replica_id * base + min(replica_id, rem)
    size = base + (1 if replica_id < rem else 0)
    return files[start:start + size]


def slice_dataset_indices(total, num_replicas, replica_id):
    """Index-range slice for in-memory datasets (per-replica sampler)."""
    base = total // num_replicas
    rem = total % num_replicas
    start = replica_id * base + min(replica_id, rem)
    size = base + (1 if replica_id < rem else 0)
    return range(start, start + size)
