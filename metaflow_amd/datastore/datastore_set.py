"""Concurrent prefetch of many task datastores.

Parity target: /root/reference/metaflow/datastore/datastore_set.py:15
(TaskDataStoreSet) — used by joins with many inputs and by resume to avoid
serial datastore init.
"""

from concurrent.futures import ThreadPoolExecutor


class TaskDataStoreSet(object):
    def __init__(self, flow_datastore, run_id, steps=None,
                 prefetch_data_artifacts=None, max_workers=8):
        self._fds = flow_datastore
        self.run_id = run_id
        specs = []
        for step in (steps or flow_datastore.list_steps(run_id)):
            for task_id in flow_datastore.list_tasks(run_id, step):
                specs.append((step, task_id))

        def load(spec):
            step, task_id = spec
            ds = self._fds.get_task_datastore(run_id, step, task_id)
            if prefetch_data_artifacts and ds.attempt is not None:
                names = [n for n in prefetch_data_artifacts if n in ds]
                for _ in ds.load_artifacts(names):
                    pass
            return (step, task_id), ds

        self._stores = {}
        with ThreadPoolExecutor(max_workers=max_workers) as pool:
            for key, ds in pool.map(load, specs):
                self._stores[key] = ds

    def get_with_pathspec(self, step, task_id):
        return self._stores.get((step, str(task_id)))

    def __iter__(self):
        return iter(self._stores.values())

    def __len__(self):
        return len(self._stores)
