from .storage import DataStoreStorage, LocalStorage, STORAGE_IMPLS
from .cas import ContentAddressedStore
from .task_datastore import TaskDataStore
from .flow_datastore import FlowDataStore

__all__ = [
    "DataStoreStorage",
    "LocalStorage",
    "STORAGE_IMPLS",
    "ContentAddressedStore",
    "TaskDataStore",
    "FlowDataStore",
]
