"""Spark Estimator API (reference: horovod/spark/common/estimator.py:25-98,
torch/estimator.py): fit a torch model on a Spark DataFrame with
data-parallel horovod_amd training, returning a transformer for inference.

Two data paths:
 * with a Store (reference: Store + petastorm): the DataFrame is
   materialized to parquet under the store prefix and each rank streams its
   strided row-group shard via pyarrow (data/parquet_loader.py) — no rank
   ever holds the whole dataset;
 * without a Store: the legacy collect() path for toy frames.

`fit_on_parquet` (reference: estimator.py fit_on_parquet) skips Spark
entirely and trains local/ssh workers straight off a parquet dataset.
"""
import copy
import os
import sys
import tempfile

import cloudpickle
import torch

import horovod_amd.spark as hvd_spark
from horovod_amd.spark.store import Store


class TorchEstimator:
    def __init__(self, model=None, optimizer=None, loss=None,
                 feature_cols=None, label_cols=None, batch_size=32,
                 epochs=1, num_proc=None, backward_passes_per_step=1,
                 store=None, run_id=None, verbose=1):
        self.model = model
        self.optimizer = optimizer
        self.loss = loss
        self.feature_cols = feature_cols or ["features"]
        self.label_cols = label_cols or ["label"]
        self.batch_size = batch_size
        self.epochs = epochs
        self.num_proc = num_proc
        self.backward_passes_per_step = backward_passes_per_step
        self.store = store
        self.run_id = run_id
        self.verbose = verbose

    # -- parquet path (no Spark required) ----------------------------------
    def _make_spec(self, train_path, store, checkpoint_path):
        opt = self.optimizer
        opt_state = opt.state_dict() if opt is not None else None
        opt_cls = type(opt) if opt is not None else None
        lr = (opt.param_groups[0].get("lr", 0.01)
              if opt is not None else 0.01)

        def opt_factory(params, _cls=opt_cls, _lr=lr, _state=opt_state):
            if _cls is None:
                return torch.optim.SGD(params, lr=_lr)
            o = _cls(params, lr=_lr)
            if _state:
                try:
                    o.load_state_dict(_state)
                except Exception:
                    pass
            return o

        return {
            "model": copy.deepcopy(self.model),
            "opt_factory": opt_factory,
            "loss": self.loss,
            "feature_cols": self.feature_cols,
            "label_cols": self.label_cols,
            "batch_size": self.batch_size,
            "epochs": self.epochs,
            "backward_passes_per_step": self.backward_passes_per_step,
            "train_path": train_path,
            "store": store,
            "checkpoint_path": checkpoint_path,
        }

    def fit_on_parquet(self, train_path, num_proc=None, store=None):
        """Train directly on a parquet dataset with local workers
        (reference: estimator.py fit_on_parquet via petastorm; here via
        pyarrow row-group shards + the slot-env launcher)."""
        from horovod_amd.runner.launch import run_command_local

        store = store or self.store or Store.create(
            tempfile.mkdtemp(prefix="hvd_est_"))
        run_id = self.run_id or store.new_run_id()
        ckpt = store.get_checkpoint_path(run_id)
        spec = self._make_spec(train_path, store, ckpt)
        np_ = num_proc or self.num_proc or 1
        with tempfile.NamedTemporaryFile(suffix=".pkl", delete=False) as f:
            cloudpickle.dump(spec, f)
            spec_path = f.name
        try:
            codes = run_command_local(
                np_, [sys.executable, "-m", "horovod_amd.spark._fit_task",
                      spec_path])
            if any(codes):
                raise RuntimeError(
                    f"fit_on_parquet workers failed (exit codes {codes})")
        finally:
            os.unlink(spec_path)
        import io
        state = torch.load(io.BytesIO(store.read(ckpt)), weights_only=True)
        fitted = copy.deepcopy(self.model)
        fitted.load_state_dict(state)
        return TorchModel(fitted, self.feature_cols)

    def fit(self, df):
        if self.store is not None:
            return self._fit_via_store(df)
        return self._fit_collect(df)

    def _fit_via_store(self, df):
        """Materialize the DataFrame as parquet under the store, then train
        Spark barrier tasks off pyarrow shards (reference: prepare_data +
        petastorm readers)."""
        store = self.store
        run_id = self.run_id or store.new_run_id()
        train_path = store.get_train_data_path(run_id)
        df.select(*self.feature_cols, *self.label_cols) \
          .write.mode("overwrite").parquet(train_path)
        ckpt = store.get_checkpoint_path(run_id)
        spec = self._make_spec(train_path, store, ckpt)
        blob = cloudpickle.dumps(spec)

        def train():
            import io
            import horovod_amd.torch as hvd
            from horovod_amd.data.parquet_loader import ParquetShardDataset
            sp = cloudpickle.loads(blob)
            hvd.init()
            m = sp["model"]
            opt = sp["opt_factory"](m.parameters())
            opt = hvd.DistributedOptimizer(
                opt, named_parameters=m.named_parameters(),
                backward_passes_per_step=sp["backward_passes_per_step"])
            hvd.broadcast_parameters(m.state_dict(), root_rank=0)
            loss_fn = sp["loss"] or torch.nn.functional.mse_loss
            for _ in range(sp["epochs"]):
                loader = ParquetShardDataset(
                    sp["train_path"], sp["feature_cols"], sp["label_cols"],
                    batch_size=sp["batch_size"], rank=hvd.rank(),
                    size=hvd.size())
                for x, y in loader:
                    opt.zero_grad()
                    loss_fn(m(x), y).backward()
                    opt.step()
            hvd.join()  # uneven shard batch counts (see _fit_task)
            if hvd.rank() == 0:
                buf = io.BytesIO()
                torch.save(m.state_dict(), buf)
                sp["store"].write(sp["checkpoint_path"], buf.getvalue())
            return None

        hvd_spark.run(train, num_proc=self.num_proc)
        import io
        state = torch.load(io.BytesIO(store.read(ckpt)), weights_only=True)
        fitted = copy.deepcopy(self.model)
        fitted.load_state_dict(state)
        return TorchModel(fitted, self.feature_cols)

    def _fit_collect(self, df):
        model = copy.deepcopy(self.model)
        opt_state = self.optimizer.state_dict() if self.optimizer else None
        loss_fn = self.loss or torch.nn.functional.mse_loss
        feature_cols, label_cols = self.feature_cols, self.label_cols
        batch_size, epochs = self.batch_size, self.epochs
        bpps = self.backward_passes_per_step

        rows = df.select(*feature_cols, *label_cols).collect()
        feats = torch.tensor([[float(v) for c in feature_cols
                               for v in _as_seq(r[c])] for r in rows])
        labels = torch.tensor([[float(v) for c in label_cols
                                for v in _as_seq(r[c])] for r in rows])

        def train():
            import horovod_amd.torch as hvd
            hvd.init()
            m = copy.deepcopy(model)
            opt = torch.optim.SGD(m.parameters(), lr=0.01)
            if opt_state:
                try:
                    opt.load_state_dict(opt_state)
                except Exception:
                    pass
            opt = hvd.DistributedOptimizer(
                opt, named_parameters=m.named_parameters(),
                backward_passes_per_step=bpps)
            hvd.broadcast_parameters(m.state_dict(), root_rank=0)
            n = feats.shape[0]
            shard = slice(hvd.rank(), n, hvd.size())
            x, y = feats[shard], labels[shard]
            for _ in range(epochs):
                for i in range(0, len(x), batch_size):
                    opt.zero_grad()
                    out = m(x[i:i + batch_size])
                    l = loss_fn(out, y[i:i + batch_size])
                    l.backward()
                    opt.step()
            return m.state_dict() if hvd.rank() == 0 else None

        results = hvd_spark.run(train, num_proc=self.num_proc)
        state = next(r for r in results if r is not None)
        fitted = copy.deepcopy(self.model)
        fitted.load_state_dict(state)
        return TorchModel(fitted, self.feature_cols)


def _as_seq(v):
    try:
        return list(v)
    except TypeError:
        return [v]


class TorchModel:
    """Transformer returned by TorchEstimator.fit (reference:
    HorovodModel.transform)."""

    def __init__(self, model, feature_cols):
        self.model = model.eval()
        self.feature_cols = feature_cols

    def transform(self, df):
        import pyspark.sql.functions as F
        from pyspark.sql.types import ArrayType, DoubleType

        model = self.model
        feature_cols = self.feature_cols

        @F.udf(ArrayType(DoubleType()))
        def predict(*cols):
            x = torch.tensor([[float(v) for c in cols for v in _as_seq(c)]])
            with torch.no_grad():
                return [float(v) for v in model(x).flatten()]

        return df.withColumn("prediction", predict(*feature_cols))
