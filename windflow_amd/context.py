"""RuntimeContext analog (reference wf/context.hpp:53, local_storage.hpp:57).

Per-batch Python callbacks receive the runtime context inline in the
column dict: `cols['replica']`, `cols['parallelism']`, `cols['watermark']`,
`cols['stream_tag']` (joins) — the engine fills these per call (see
bindings.cpp batch_views).  Source callbacks receive (replica,
parallelism) positionally.

LocalStorage mirrors the reference's name→object per-replica store.  With
Python closures a plain dict usually suffices; `per_replica` helps when one
callable is shared by several replicas:

    store = LocalStorage()
    def fn(cols):
        mine = store.per_replica(cols['replica'])
        mine['count'] = mine.get('count', 0) + len(cols['ts'])
"""


class LocalStorage(dict):
    """name -> object store (reference LocalStorage: isContained/get/put)."""

    def per_replica(self, replica):
        """A private sub-dict per replica index (one callable, n replicas)."""
        return self.setdefault(('__replica__', replica), {})

    def isContained(self, name):
        return name in self

    def get_obj(self, name, default=None):
        return super().get(name, default)

    def put(self, name, obj):
        self[name] = obj
        return obj
