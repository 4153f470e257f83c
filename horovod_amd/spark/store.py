"""Storage abstraction for Estimator intermediates and checkpoints.

Reference: horovod/spark/common/store.py:38-540 — Store / FilesystemStore /
LocalStore / HDFSStore / DBFSLocalStore manage the layout of intermediate
training data (parquet), checkpoints and logs, and hand out filesystem
accessors to the training processes.

MI355X-native design: pyarrow.fs is the single filesystem layer (installed
in the image; it covers local + HDFS without a pyspark dependency), and the
per-run layout matches the reference so tooling that expects
`<prefix>/intermediate_train_data` etc. keeps working.
"""
import os
import uuid


class Store:
    """Layout + IO for one training prefix (reference store.py:38-114)."""

    def __init__(self):
        self._train_path = None
        self._val_path = None
        self._test_path = None

    @staticmethod
    def create(prefix_path, *args, **kwargs):
        """Pick a store from the path scheme (reference store.py:100-114)."""
        if prefix_path.startswith("hdfs://"):
            return HDFSStore(prefix_path, *args, **kwargs)
        if prefix_path.startswith("dbfs:/") or \
                prefix_path.startswith("/dbfs"):
            return DBFSLocalStore(prefix_path, *args, **kwargs)
        return LocalStore(prefix_path, *args, **kwargs)

    # -- layout (reference store.py:116-170) -------------------------------
    def get_train_data_path(self, idx=None):
        raise NotImplementedError()

    def get_val_data_path(self, idx=None):
        raise NotImplementedError()

    def get_test_data_path(self, idx=None):
        raise NotImplementedError()

    def get_data_metadata_path(self, path):
        return os.path.join(path, "_metadata.pkl")

    def get_checkpoint_path(self, run_id):
        raise NotImplementedError()

    def get_logs_path(self, run_id):
        raise NotImplementedError()

    def get_checkpoint_filename(self):
        return "checkpoint.pt"

    def get_logs_subdir(self):
        return "logs"

    def exists(self, path):
        raise NotImplementedError()

    def read(self, path):
        raise NotImplementedError()

    def write(self, path, data):
        raise NotImplementedError()

    def is_parquet_dataset(self, path):
        raise NotImplementedError()

    def saving_runs(self):
        """Whether checkpoints/logs are persisted (reference
        store.py:92-98)."""
        raise NotImplementedError()


class AbstractFilesystemStore(Store):
    """Store over a pyarrow filesystem (reference FilesystemStore,
    store.py:172-337)."""

    def __init__(self, prefix_path, train_path=None, val_path=None,
                 test_path=None, runs_path=None, save_runs=True):
        super().__init__()
        self.prefix_path = prefix_path.rstrip("/")
        self._train_path = train_path or self._default("intermediate_train_data")
        self._val_path = val_path or self._default("intermediate_val_data")
        self._test_path = test_path or self._default("intermediate_test_data")
        self._runs_path = runs_path or self._default("runs")
        self._save_runs = save_runs

    def _default(self, leaf):
        return f"{self.prefix_path}/{leaf}"

    def _fs_path(self, path):
        """Return (pyarrow_filesystem, fs-local path)."""
        raise NotImplementedError()

    def get_train_data_path(self, idx=None):
        return self._train_path if idx is None else \
            f"{self._train_path}.{idx}"

    def get_val_data_path(self, idx=None):
        return self._val_path if idx is None else f"{self._val_path}.{idx}"

    def get_test_data_path(self, idx=None):
        return self._test_path if idx is None else f"{self._test_path}.{idx}"

    def get_run_path(self, run_id):
        return f"{self._runs_path}/{run_id}"

    def get_checkpoint_path(self, run_id):
        return f"{self.get_run_path(run_id)}/{self.get_checkpoint_filename()}" \
            if self._save_runs else None

    def get_logs_path(self, run_id):
        return f"{self.get_run_path(run_id)}/{self.get_logs_subdir()}" \
            if self._save_runs else None

    def saving_runs(self):
        return self._save_runs

    def exists(self, path):
        import pyarrow.fs as pafs
        fs, p = self._fs_path(path)
        info = fs.get_file_info(p)
        return info.type != pafs.FileType.NotFound

    def read(self, path):
        fs, p = self._fs_path(path)
        with fs.open_input_stream(p) as f:
            return f.read()

    def write(self, path, data):
        fs, p = self._fs_path(path)
        parent = p.rsplit("/", 1)[0] if "/" in p else ""
        if parent:
            fs.create_dir(parent, recursive=True)
        with fs.open_output_stream(p) as f:
            f.write(data)

    def is_parquet_dataset(self, path):
        try:
            import pyarrow.parquet as pq
            fs, p = self._fs_path(path)
            pq.ParquetDataset(p, filesystem=fs)
            return True
        except Exception:
            return False

    def new_run_id(self):
        return f"run_{uuid.uuid4().hex[:12]}"


class LocalStore(AbstractFilesystemStore):
    """Local-filesystem store (reference store.py:339-360)."""

    def _fs_path(self, path):
        import pyarrow.fs as pafs
        p = path
        if p.startswith("file://"):
            p = p[len("file://"):]
        return pafs.LocalFileSystem(), p


class DBFSLocalStore(LocalStore):
    """Databricks DBFS paths exposed through the local FUSE mount
    (reference store.py:499-540): `dbfs:/...` -> `/dbfs/...`."""

    def __init__(self, prefix_path, *args, **kwargs):
        if prefix_path.startswith("dbfs:/"):
            prefix_path = "/dbfs/" + prefix_path[len("dbfs:/"):].lstrip("/")
        super().__init__(prefix_path, *args, **kwargs)

    def get_checkpoint_filename(self):
        # TF-format quirk in the reference; torch checkpoints keep .pt
        return "checkpoint.pt"


class HDFSStore(AbstractFilesystemStore):
    """HDFS store via pyarrow's libhdfs binding (reference
    store.py:362-497).  Requires a reachable namenode + libhdfs at runtime;
    construction is lazy so unit tests can exercise the layout."""

    def __init__(self, prefix_path, host=None, port=None, user=None,
                 **kwargs):
        self._host = host
        self._port = port
        self._user = user
        super().__init__(prefix_path, **kwargs)
        self._hdfs = None

    def _connect(self):
        if self._hdfs is None:
            import pyarrow.fs as pafs
            self._hdfs = pafs.HadoopFileSystem(
                host=self._host or "default", port=self._port or 0,
                user=self._user)
        return self._hdfs

    def _fs_path(self, path):
        p = path
        if p.startswith("hdfs://"):
            p = "/" + p[len("hdfs://"):].split("/", 1)[1]
        return self._connect(), p
