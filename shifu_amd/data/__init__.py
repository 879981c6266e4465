from shifu_amd.data.csv_loader import TabularDataset, load_csv_files  # noqa: F401
from shifu_amd.data.synthetic import generate_synthetic_csv, synthetic_tensors  # noqa: F401
from shifu_amd.data.sharding import shard_files, shard_rows  # noqa: F401
