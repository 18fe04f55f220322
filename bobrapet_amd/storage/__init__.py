from .manager import StorageManager, extract_path, parse_path, PathError, RefError, STORAGE_REF_KEY  # noqa: F401
from .stores import BlobNotFound, FileStore, MemStore, Store, TensorStore  # noqa: F401
