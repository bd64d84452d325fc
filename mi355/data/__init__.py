from .dataset import (ActorPoolStrategy, Dataset, DatasetPipeline,
                      GroupedData)
from .preprocessor import (BatchMapper, Chain, LabelEncoder, MinMaxScaler,
                           PowerTransformer, Preprocessor, StandardScaler)

from_items = Dataset.from_items
from_pandas = Dataset.from_pandas
from_numpy = Dataset.from_numpy
from_arrow = Dataset.from_arrow
from_huggingface = Dataset.from_huggingface
read_parquet = Dataset.read_parquet
read_csv = Dataset.read_csv

__all__ = [
    "Dataset", "DatasetPipeline", "GroupedData", "ActorPoolStrategy", "Preprocessor",
    "BatchMapper", "MinMaxScaler", "StandardScaler", "PowerTransformer",
    "LabelEncoder", "Chain", "from_items", "from_pandas", "from_numpy",
    "from_arrow", "from_huggingface", "read_parquet", "read_csv",
]
