"""Parquet-shard dataset: the petastorm-equivalent data path.

Reference: horovod/spark uses a Store + petastorm to feed materialized
parquet to training processes (spark/common/store.py + data loaders).
MI355X-native design: pyarrow reads the parquet row groups directly (no
petastorm dependency); each rank owns a strided shard of row groups so no
rank materializes the whole dataset.
"""
import torch

from horovod_amd.data.data_loader_base import (AsyncDataLoaderMixin,
                                               BaseDataLoader)


class ParquetShardDataset(BaseDataLoader):
    """Iterates (features, labels) minibatches from this rank's shard of a
    parquet dataset.  Row groups are strided across ranks (rank, size), so
    IO is disjoint and memory is bounded by one row group."""

    def __init__(self, path, feature_cols, label_cols, batch_size=32,
                 rank=0, size=1, filesystem=None, dtype=torch.float32):
        import pyarrow.parquet as pq
        self._pf = pq.ParquetFile(path) if not _is_dir(path, filesystem) \
            else None
        self._dataset_path = path
        self._filesystem = filesystem
        self.feature_cols = list(feature_cols)
        self.label_cols = list(label_cols)
        self.batch_size = batch_size
        self.rank = rank
        self.size = size
        self.dtype = dtype

    def _fragments(self):
        import pyarrow.parquet as pq
        if self._pf is not None:
            for g in range(self._pf.num_row_groups):
                if g % self.size == self.rank:
                    yield self._pf.read_row_group(
                        g, columns=self.feature_cols + self.label_cols)
        else:
            ds = pq.ParquetDataset(self._dataset_path,
                                   filesystem=self._filesystem)
            for i, frag in enumerate(ds.fragments):
                if i % self.size == self.rank:
                    yield frag.to_table(
                        columns=self.feature_cols + self.label_cols)

    @staticmethod
    def _to_tensor(table, cols, dtype):
        import numpy as np
        mats = []
        for c in cols:
            col = table.column(c).to_pylist()
            arr = np.asarray(
                [v if isinstance(v, (list, tuple)) else [v] for v in col],
                dtype="float64")
            mats.append(arr)
        return torch.from_numpy(np.concatenate(mats, axis=1)).to(dtype)

    def _iterate(self):
        for table in self._fragments():
            x = self._to_tensor(table, self.feature_cols, self.dtype)
            y = self._to_tensor(table, self.label_cols, self.dtype)
            for i in range(0, len(x), self.batch_size):
                yield x[i:i + self.batch_size], y[i:i + self.batch_size]


def _is_dir(path, filesystem):
    import os
    if filesystem is None:
        return os.path.isdir(path)
    import pyarrow.fs as pafs
    return filesystem.get_file_info(path).type == pafs.FileType.Directory


class AsyncParquetShardLoader(AsyncDataLoaderMixin, ParquetShardDataset):
    """Background-thread prefetching variant (reference: petastorm async
    loader via AsyncDataLoaderMixin)."""
