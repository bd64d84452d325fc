from mi355.data.preprocessor import (BatchMapper, Chain, LabelEncoder,
                                     MinMaxScaler, PowerTransformer,
                                     Preprocessor, StandardScaler)

__all__ = ["Preprocessor", "BatchMapper", "MinMaxScaler", "StandardScaler",
           "PowerTransformer", "LabelEncoder", "Chain"]
