from . import avro_io, model_io

__all__ = ["avro_io", "model_io"]
