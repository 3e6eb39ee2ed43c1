from .export import export_onnx_model

__all__ = ["export_onnx_model"]
