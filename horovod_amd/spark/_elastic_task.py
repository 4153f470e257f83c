"""Worker bootstrap for horovod_amd.spark.run_elastic: loads the pickled
user function and runs it under the elastic worker protocol (env provided
by the ElasticDriver).  Rank 0 persists the result for the driver."""
import pickle
import sys

import cloudpickle


def main(fn_path, out_path):
    with open(fn_path, "rb") as f:
        fn, args, kwargs = cloudpickle.load(f)
    import horovod_amd.torch as hvd
    hvd.init()
    result = fn(*args, **kwargs)
    if hvd.rank() == 0:
        with open(out_path, "wb") as f:
            pickle.dump(result, f)
    hvd.shutdown()


if __name__ == "__main__":
    main(sys.argv[1], sys.argv[2])
