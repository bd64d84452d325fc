from mi355.data import (ActorPoolStrategy, Dataset, from_arrow,
                        from_huggingface, from_items, from_numpy, from_pandas,
                        read_csv, read_parquet)
from . import preprocessors  # noqa: F401
from mi355.data.preprocessor import (BatchMapper, Chain, LabelEncoder,
                                     MinMaxScaler, PowerTransformer,
                                     Preprocessor, StandardScaler)

__all__ = ["Dataset", "ActorPoolStrategy", "from_items", "from_pandas",
           "from_numpy", "from_arrow", "from_huggingface", "read_parquet",
           "read_csv", "preprocessors", "Preprocessor", "BatchMapper",
           "MinMaxScaler", "StandardScaler", "PowerTransformer",
           "LabelEncoder", "Chain"]
