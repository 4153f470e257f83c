"""Spark/Ray integration depth tests (round 2).

pyspark and ray are not in the image, so coverage follows the reference's
unit-test strategy (test/single/test_ray*.py uses fakes): the Store
hierarchy, the parquet data path and the estimator's no-Spark
fit_on_parquet run for real; Ray discovery runs against a fake ray module;
the elastic executor machinery runs with an injected discovery and real
worker processes.
"""
import os
import sys
import types

import pytest
import torch

from tests.parallel_util import REPO


@pytest.fixture()
def parquet_xy(tmp_path):
    import numpy as np
    import pyarrow as pa
    import pyarrow.parquet as pq
    rng = np.random.default_rng(3)
    x = rng.normal(size=(240, 4))
    w = np.array([1.0, -2.0, 0.5, 3.0])
    y = x @ w
    t = pa.table({"features": [list(r) for r in x],
                  "label": [[float(v)] for v in y]})
    path = str(tmp_path / "train.parquet")
    pq.write_table(t, path, row_group_size=24)
    return path


def test_store_hierarchy(tmp_path):
    from horovod_amd.spark.store import (DBFSLocalStore, HDFSStore,
                                         LocalStore, Store)
    prefix = str(tmp_path / "store")
    s = Store.create(prefix)
    assert isinstance(s, LocalStore)
    assert s.get_train_data_path().endswith("intermediate_train_data")
    assert s.get_train_data_path(3).endswith("intermediate_train_data.3")
    run = s.new_run_id()
    ckpt = s.get_checkpoint_path(run)
    assert run in ckpt and ckpt.endswith("checkpoint.pt")
    assert not s.exists(ckpt)
    s.write(ckpt, b"hello-checkpoint")
    assert s.exists(ckpt)
    assert s.read(ckpt) == b"hello-checkpoint"
    assert s.saving_runs()
    # scheme dispatch
    assert isinstance(Store.create("hdfs://nn:9000/x"), HDFSStore)
    d = Store.create("dbfs:/ml/prefix")
    assert isinstance(d, DBFSLocalStore)
    assert d.prefix_path.startswith("/dbfs/")
    # logs layout
    assert s.get_logs_path(run).endswith("logs")


def test_store_parquet_detection(tmp_path, parquet_xy):
    from horovod_amd.spark.store import Store
    s = Store.create(str(tmp_path))
    assert s.is_parquet_dataset(parquet_xy)
    assert not s.is_parquet_dataset(str(tmp_path / "nope"))


def test_fit_on_parquet_np2(tmp_path, parquet_xy):
    """End-to-end estimator without Spark: 2 local workers train a linear
    model off disjoint parquet shards; the checkpoint round-trips through
    the Store and the fitted model beats the init loss by a wide margin."""
    from horovod_amd.spark.estimator import TorchEstimator
    from horovod_amd.spark.store import Store

    os.environ.setdefault("PYTHONPATH", "")
    os.environ["PYTHONPATH"] = REPO + os.pathsep + os.environ["PYTHONPATH"]
    torch.manual_seed(0)
    model = torch.nn.Linear(4, 1)
    est = TorchEstimator(model=model,
                         optimizer=torch.optim.SGD(model.parameters(),
                                                   lr=0.05),
                         feature_cols=["features"], label_cols=["label"],
                         batch_size=16, epochs=8,
                         store=Store.create(str(tmp_path / "st")))
    fitted = est.fit_on_parquet(parquet_xy, num_proc=2)

    import numpy as np
    rng = np.random.default_rng(3)
    x = rng.normal(size=(240, 4))
    w = np.array([1.0, -2.0, 0.5, 3.0])
    xt = torch.tensor(x, dtype=torch.float32)
    yt = torch.tensor(x @ w, dtype=torch.float32).unsqueeze(1)
    with torch.no_grad():
        fit_loss = torch.nn.functional.mse_loss(fitted.model(xt), yt).item()
        init_loss = torch.nn.functional.mse_loss(model(xt), yt).item()
    assert fit_loss < init_loss * 0.2, (fit_loss, init_loss)


def test_ray_host_discovery_fake():
    """RayHostDiscovery against a fake ray module (reference test model:
    test/single/test_ray_elastic uses fake discovery/clusters)."""
    fake = types.ModuleType("ray")
    fake.nodes = lambda: [
        {"Alive": True, "NodeManagerAddress": "10.0.0.1",
         "Resources": {"CPU": 8.0, "GPU": 4.0}},
        {"Alive": True, "NodeManagerAddress": "10.0.0.2",
         "Resources": {"CPU": 16.0}},
        {"Alive": False, "NodeManagerAddress": "10.0.0.3",
         "Resources": {"CPU": 8.0, "GPU": 8.0}},
    ]
    sys.modules["ray"] = fake
    try:
        from horovod_amd.ray import RayHostDiscovery
        gpu_disc = RayHostDiscovery(use_gpu=True)
        assert gpu_disc.find_available_hosts_and_slots() == {"10.0.0.1": 4}
        cpu_disc = RayHostDiscovery(cpus_per_worker=4)
        assert cpu_disc.find_available_hosts_and_slots() == {
            "10.0.0.1": 2, "10.0.0.2": 4}
    finally:
        del sys.modules["ray"]


def test_elastic_ray_executor_with_injected_discovery(tmp_path):
    """ElasticRayExecutor end-to-end with an injected discovery (no ray):
    exercises the same path run_elastic uses — pickled fn shipped to real
    worker processes under the elastic rendezvous, result returned."""
    from horovod_amd.ray import ElasticRayExecutor
    from horovod_amd.runner.elastic_driver import FixedHosts

    def train_fn():
        import horovod_amd.torch as hvd
        t = hvd.allreduce(torch.ones(4), average=False, name="rayex")
        return float(t[0].item())

    old = os.environ.get("PYTHONPATH", "")
    os.environ["PYTHONPATH"] = REPO + (os.pathsep + old if old else "")
    try:
        ex = ElasticRayExecutor(
            override_discovery=FixedHosts({"127.0.0.1": 2}), min_np=2)
        ex.start()
        result = ex.run(train_fn)
    finally:
        os.environ["PYTHONPATH"] = old
    assert result == 2.0, result
