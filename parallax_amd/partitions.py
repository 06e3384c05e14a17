"""Sparse-variable partitioning + the partition-count auto-search.

MI355X-native equivalent of the reference's common/partitions.py (170 loc).

``get_partitioner(min_num_partitions)`` keeps the reference contract
(partitions.py:35-51): the partition count is taken from the
``PARALLAX_PARTITIONS`` env var when the search harness set one, else the
user's minimum; the returned object row-partitions a large embedding into
``p`` fixed-size shards (our PartitionedEmbedding consumes it).

``PartitionStatCollector`` reproduces the doubling/halving search driven by
measured iteration time and the cost-model fit
``t(n) = b/n + a(n-1) + c`` via scipy curve_fit (partitions.py:74-170).
The measurement transport is a local multiprocessing.Manager queue instead
of a remote BaseManager TCP queue — single-node launch plane.
"""

import os
import time

import numpy as np

from parallax_amd.consts import (PARALLAX_MIN_PARTITIONS,
                                 PARALLAX_PARTITIONS)
from parallax_amd.lib import parallax_log


class FixedSizePartitioner(object):
    """Row partitioner: splits dim-0 of a variable into ``num_partitions``
    nearly equal shards (reference tf.fixed_size_partitioner semantics:
    first ``total % p`` shards get one extra row)."""

    def __init__(self, num_partitions):
        assert num_partitions >= 1
        self.num_partitions = int(num_partitions)

    def shard_sizes(self, total_rows):
        p = min(self.num_partitions, total_rows)
        base = total_rows // p
        extra = total_rows % p
        return [base + (1 if i < extra else 0) for i in range(p)]

    def shard_offsets(self, total_rows):
        sizes = self.shard_sizes(total_rows)
        offs, acc = [], 0
        for s in sizes:
            offs.append(acc)
            acc += s
        return offs

    def __call__(self, total_rows):
        return list(zip(self.shard_offsets(total_rows),
                        self.shard_sizes(total_rows)))


def get_partitioner(min_num_partitions=1):
    """Return a FixedSizePartitioner with a count determined by Parallax
    (reference partitions.py:35-51: env-var override from the search)."""
    if PARALLAX_MIN_PARTITIONS not in os.environ:
        os.environ[PARALLAX_MIN_PARTITIONS] = str(min_num_partitions)
    if PARALLAX_PARTITIONS in os.environ:
        partitions = int(os.environ[PARALLAX_PARTITIONS])
    else:
        partitions = min_num_partitions
    return FixedSizePartitioner(partitions)


def fit_cost_model(p_list, exec_time_list):
    """Fit t(n) = b/n + a(n-1) + c and return argmin over [min_p, max_p]
    (reference partitions.py:140-170 ``_find_optimal_p``)."""
    if len(p_list) < 3:
        return p_list[int(np.argmin(exec_time_list))]
    from scipy import optimize
    max_time = float(max(exec_time_list))
    exec_times = [t / max_time for t in exec_time_list]

    def fitfunc(n, a, b, c):
        return b / n + a * (n - 1) + c

    popt, _ = optimize.curve_fit(fitfunc, np.array(p_list, dtype=float),
                                 np.array(exec_times, dtype=float))
    min_p, max_p = int(min(p_list)), int(max(p_list))
    best_p, best_t = None, None
    for i in range(min_p, max_p + 1):
        pred = fitfunc(i, *popt)
        if best_t is None or pred < best_t:
            best_t, best_p = pred, i
    return best_p


class PartitionStatCollector(object):
    """Doubling/halving walk over the partition count, driven by measured
    mean iteration time, then cost-model argmin
    (reference partitions.py:74-138 ``recv_exec_time``).

    The walk logic is factored into the pure method ``update(exec_time,
    alive)`` -> (search_continues, next_p) so it is directly unit-testable;
    the process-management wrapper lives in runner.py.
    """

    def __init__(self, p_to_test, min_partitions=None, max_trials=12):
        self.p_to_test = p_to_test
        self.prev_p = None
        self.prev_exec_time = None
        self.exec_time_list = []
        self.p_list = []
        self.start = None
        self.max_trials = max_trials
        if min_partitions is None:
            min_partitions = int(os.environ.get(PARALLAX_MIN_PARTITIONS, '1'))
        self.min_partitions = min_partitions

    def update(self, exec_time, alive=True):
        """One search step.  ``exec_time`` = mean measured iteration time for
        the partition count just tested (None when the job died —
        reference partitions.py:122-128 treats death as 'p too small')."""
        if self.start is None:
            self.start = time.time()
        stop = False
        if alive and exec_time is not None:
            curr_p = self.p_to_test
            self.p_list.append(curr_p)
            self.exec_time_list.append(exec_time)
            if self.prev_p:
                if self.prev_exec_time < exec_time:
                    if self.prev_p > curr_p:
                        stop = True
                    else:
                        self.p_to_test = min(self.p_list) // 2
                else:
                    if self.prev_p < curr_p:
                        self.p_to_test *= 2
                    else:
                        self.p_to_test //= 2
                if self.p_to_test < self.min_partitions:
                    stop = True
            else:
                self.p_to_test *= 2
            if len(self.p_list) >= self.max_trials:
                stop = True  # noise guard beyond the reference walk
            self.prev_p = curr_p
            self.prev_exec_time = exec_time
        else:
            # job died: p too small (reference partitions.py:122-128)
            self._deaths = getattr(self, '_deaths', 0) + 1
            if self.prev_p:
                stop = True
            elif self._deaths >= self.max_trials:
                # every trial died before a single measurement: a
                # persistent failure, not an OOM walk — stop doubling
                raise RuntimeError(
                    'partition search: %d consecutive job deaths with no '
                    'successful trial (last p=%d) — the failure is not '
                    'partition-related' % (self._deaths, self.p_to_test))
            else:
                self.p_to_test *= 2
                self.min_partitions = self.p_to_test

        if stop:
            self.p_to_test = fit_cost_model(self.p_list, self.exec_time_list)
            parallax_log.info(
                'optimal partitions: %d, search time: %d secs'
                % (self.p_to_test, time.time() - self.start))
        return (not stop), self.p_to_test
