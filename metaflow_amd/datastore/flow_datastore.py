"""Per-flow datastore: factory for TaskDataStores + raw blob API.

Parity target: /root/reference/metaflow/datastore/flow_datastore.py:13
(FlowDataStore, get_task_datastores :79, save_data/load_data for code
packages).
"""

from .cas import ContentAddressedStore
from .task_datastore import TaskDataStore


class FlowDataStore(object):
    def __init__(self, flow_name, storage):
        self.flow_name = flow_name
        self.storage = storage
        self.ca_store = ContentAddressedStore(
            storage.path_join(flow_name, "data"), storage)

    @property
    def datastore_root(self):
        return self.storage.root

    def get_task_datastore(self, run_id, step_name, task_id, attempt=None,
                           mode="r", data_metadata=None):
        return TaskDataStore(self, run_id, step_name, task_id,
                             attempt=attempt, mode=mode,
                             data_metadata=data_metadata)

    # ------------------------------------------------------------- discovery
    def list_runs(self):
        out = []
        for path, is_file in self.storage.list_content([self.flow_name]):
            name = self.storage.basename(path)
            if not is_file and name != "data" and not name.startswith("_"):
                out.append(name)
        return out

    def list_steps(self, run_id):
        prefix = self.storage.path_join(self.flow_name, str(run_id))
        return [self.storage.basename(p)
                for p, is_file in self.storage.list_content([prefix])
                if not is_file]

    def list_tasks(self, run_id, step_name):
        prefix = self.storage.path_join(self.flow_name, str(run_id),
                                        step_name)
        return [self.storage.basename(p)
                for p, is_file in self.storage.list_content([prefix])
                if not is_file]

    def get_done_task_datastores(self, run_id, step_name):
        out = []
        for task_id in self.list_tasks(run_id, step_name):
            ds = self.get_task_datastore(run_id, step_name, task_id)
            if ds.attempt is not None:
                out.append(ds)
        return out

    # ------------------------------------------------------------- raw blobs
    def save_data(self, data_iter):
        """Store raw byte blobs (e.g. code packages); returns (uri, key)s."""
        return self.ca_store.save_blobs(data_iter, raw=True)

    def load_data(self, keys):
        return list(self.ca_store.load_blobs(keys))
