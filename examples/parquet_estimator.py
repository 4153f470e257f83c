"""Train a model straight off a parquet dataset — no Spark required.

The Store + pyarrow data path that backs the Spark Estimator
(horovod_amd/spark/estimator.py) also works standalone:
`TorchEstimator.fit_on_parquet` launches local workers under the slot-env
protocol, each streaming its strided row-group shard.

    python examples/parquet_estimator.py [num_workers]
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import pyarrow as pa  # noqa: E402
import pyarrow.parquet as pq  # noqa: E402
import torch  # noqa: E402

from horovod_amd.spark.estimator import TorchEstimator  # noqa: E402
from horovod_amd.spark.store import Store  # noqa: E402


def main():
    num_proc = int(sys.argv[1]) if len(sys.argv) > 1 else 2
    workdir = tempfile.mkdtemp(prefix="hvd_parquet_")
    os.environ["PYTHONPATH"] = (
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))) +
        os.pathsep + os.environ.get("PYTHONPATH", ""))

    # synthesize a linear-regression parquet dataset
    rng = np.random.default_rng(0)
    x = rng.normal(size=(2000, 8))
    w = rng.normal(size=8)
    y = x @ w + 0.01 * rng.normal(size=2000)
    table = pa.table({"features": [list(r) for r in x],
                      "label": [[float(v)] for v in y]})
    train_path = os.path.join(workdir, "train.parquet")
    pq.write_table(table, train_path, row_group_size=128)

    model = torch.nn.Sequential(torch.nn.Linear(8, 16), torch.nn.ReLU(),
                                torch.nn.Linear(16, 1))
    est = TorchEstimator(
        model=model,
        optimizer=torch.optim.SGD(model.parameters(), lr=0.05),
        loss=torch.nn.functional.mse_loss,
        feature_cols=["features"], label_cols=["label"],
        batch_size=32, epochs=5,
        store=Store.create(os.path.join(workdir, "store")))
    fitted = est.fit_on_parquet(train_path, num_proc=num_proc)

    xt = torch.tensor(x, dtype=torch.float32)
    yt = torch.tensor(y, dtype=torch.float32).unsqueeze(1)
    with torch.no_grad():
        mse = torch.nn.functional.mse_loss(fitted.model(xt), yt).item()
    print(f"fit_on_parquet with {num_proc} workers: final mse {mse:.4f}")


if __name__ == "__main__":
    main()
