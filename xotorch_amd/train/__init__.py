from xotorch_amd.train.dataset import load_dataset, iterate_batches  # noqa: F401
