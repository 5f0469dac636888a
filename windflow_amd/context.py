"""RuntimeContext analog (reference wf/context.hpp:53).

Python callbacks are per-batch; replica identity is passed to source
callbacks as (replica, parallelism).  LocalStorage is a plain dict the
user closure can capture.
"""


class LocalStorage(dict):
    pass
