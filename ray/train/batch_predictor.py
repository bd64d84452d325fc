from mi355.predict import BatchPredictor

__all__ = ["BatchPredictor"]
