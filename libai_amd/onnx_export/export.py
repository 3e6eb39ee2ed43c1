"""ONNX export (reference: libai/onnx_export/gpt2_to_onnx.py:26-86).

Exports an eval-mode model via torch.onnx; TP/PP models must be built with
tp=1/pp=1 for export (the ONNX graph is single-device).
"""

import logging

import torch

__all__ = ["export_onnx_model"]

logger = logging.getLogger(__name__)


class _LogitsOnly(torch.nn.Module):
    def __init__(self, model):
        super().__init__()
        self.model = model

    def forward(self, input_ids):
        out = self.model(input_ids=input_ids)
        return out["prediction_scores"] if isinstance(out, dict) else out


def export_onnx_model(model, sample_input=None, output_path="model.onnx",
                      input_names=("input_ids",), output_names=("logits",),
                      dynamic_axes=None, opset=17, check=True):
    model = _LogitsOnly(model.eval())
    if sample_input is None:
        sample_input = torch.randint(0, 100, (1, 8))
    if dynamic_axes is None:
        dynamic_axes = {"input_ids": {0: "batch", 1: "seq"},
                        "logits": {0: "batch", 1: "seq"}}
    torch.onnx.export(
        model, (sample_input,), output_path, input_names=list(input_names),
        output_names=list(output_names), dynamic_axes=dynamic_axes,
        opset_version=opset, dynamo=False,
    )
    logger.info(f"exported ONNX model to {output_path}")
    if check:
        try:
            import onnx

            onnx.checker.check_model(onnx.load(output_path))
            logger.info("onnx.checker passed")
        except ImportError:
            logger.warning("onnx not installed; skipping checker")
    return output_path
