# ONNX exporter package (see exporter.py)
