from mi355.serve import json_request, pandas_read_json

__all__ = ["pandas_read_json", "json_request"]
