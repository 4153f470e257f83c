"""Worker bootstrap for TorchEstimator.fit_on_parquet: each rank trains on
its parquet shard and rank 0 writes the checkpoint to the store.

Launched as `python -m horovod_amd.spark._fit_task <spec.pkl>` under the
slot-env protocol (local subprocesses or Spark/ssh executors); the spec
carries the pickled model/optimizer config and data paths (reference
analogue: the serialized train function horovod.spark ships to executors,
spark/torch/remote.py).
"""
import sys

import cloudpickle
import torch


def main(spec_path):
    with open(spec_path, "rb") as f:
        spec = cloudpickle.load(f)

    import horovod_amd.torch as hvd
    from horovod_amd.data.parquet_loader import ParquetShardDataset

    hvd.init()
    model = spec["model"]
    opt = spec["opt_factory"](model.parameters()) if spec.get("opt_factory") \
        else torch.optim.SGD(model.parameters(), lr=0.01)
    opt = hvd.DistributedOptimizer(
        opt, named_parameters=model.named_parameters(),
        backward_passes_per_step=spec.get("backward_passes_per_step", 1))
    hvd.broadcast_parameters(model.state_dict(), root_rank=0)
    hvd.broadcast_optimizer_state(opt, root_rank=0, model=model)
    loss_fn = spec.get("loss") or torch.nn.functional.mse_loss

    for epoch in range(spec.get("epochs", 1)):
        loader = ParquetShardDataset(
            spec["train_path"], spec["feature_cols"], spec["label_cols"],
            batch_size=spec.get("batch_size", 32), rank=hvd.rank(),
            size=hvd.size())
        for x, y in loader:
            opt.zero_grad()
            loss = loss_fn(model(x), y)
            loss.backward()
            opt.step()
    # shards are row-group-strided, so batch counts can differ by one per
    # rank; join() completes the trailing allreduces with zero
    # contributions (the reference's uneven-data answer)
    hvd.join()

    if hvd.rank() == 0:
        import io
        buf = io.BytesIO()
        torch.save(model.state_dict(), buf)
        store = spec["store"]
        store.write(spec["checkpoint_path"], buf.getvalue())
    hvd.shutdown()


if __name__ == "__main__":
    main(sys.argv[1])
