"""Fork-based multicore map helpers, part of the public API.

Parity target: /root/reference/metaflow/multicore_utils.py:82,140
(parallel_map, parallel_imap_unordered). Fresh implementation on
multiprocessing.
"""

import multiprocessing
import os


def _effective_procs(max_parallel):
    n = os.cpu_count() or 1
    return max(1, min(n, max_parallel) if max_parallel else n)


def parallel_imap_unordered(func, iterable, max_parallel=None):
    """Yield func(x) for x in iterable, computed across processes, in
    completion order."""
    items = list(iterable)
    if len(items) <= 1:
        for x in items:
            yield func(x)
        return
    procs = _effective_procs(max_parallel)
    with multiprocessing.Pool(processes=min(procs, len(items))) as pool:
        for res in pool.imap_unordered(func, items):
            yield res


def parallel_map(func, iterable, max_parallel=None):
    """Ordered multicore map."""
    items = list(iterable)
    if len(items) <= 1:
        return [func(x) for x in items]
    procs = _effective_procs(max_parallel)
    with multiprocessing.Pool(processes=min(procs, len(items))) as pool:
        return pool.map(func, items)
