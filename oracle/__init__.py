"""ORACLE — TEST INFRASTRUCTURE ONLY.

CPU (numpy) restatement of the reference algorithms on the north-star path
(SURVEY.md §8c), used exclusively as the parity checker and as bench.py's
``cpu_baseline`` leg.  Nothing in the product path (modin_amd/*) may import,
call or link anything in this package; the HIP path fails loudly when the
extension is missing — it never routes through this code.

Pinning: the reference's own implementation (Modin PandasOnPython) is pure
Python and importable only in the build container (/root/reference does not
exist on the GPU box), so the oracle is pinned by committed golden vectors
generated there by oracle/make_golden.py (which runs the REAL reference
end-to-end), stored under tests/golden/.  tests/test_oracle_golden.py checks
this oracle against every vector; tests/test_gpu_parity.py checks the HIP
path against this oracle and the same vectors.
"""

from .ops import (  # noqa: F401
    binary_op,
    compare_op,
    filter_rows,
    groupby_agg,
    inner_join,
    map_op,
    partitioned_groupby_agg,
    pick_splitters,
    rand_cdf,
    rand_f64,
    rand_int,
    reduce_op,
    shuffle_dest,
    sort_perm,
    split_row_counts,
    zipf_cdf,
)
