"""ONNX export (reference: libai/onnx_export/gpt2_to_onnx.py:26-86 and
t5_to_onnx.py — both exporters, with onnxruntime output verification).

Exports an eval-mode model via torch.onnx; TP/PP models must be built with
tp=1/pp=1 for export (the ONNX graph is single-device).
"""

import logging

import torch

__all__ = ["export_onnx_model", "export_t5_onnx_model", "verify_onnx_model"]

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


class _T5LogitsOnly(torch.nn.Module):
    def __init__(self, model):
        super().__init__()
        self.model = model

    def forward(self, encoder_input_ids, decoder_input_ids):
        out = self.model(encoder_input_ids=encoder_input_ids,
                         decoder_input_ids=decoder_input_ids)
        return out["prediction_scores"] if isinstance(out, dict) else out


def export_t5_onnx_model(model, enc_ids=None, dec_ids=None,
                         output_path="t5.onnx", opset=17, check=True):
    """Encoder-decoder export: (enc ids, dec ids) -> decoder logits
    (reference: libai/onnx_export/t5_to_onnx.py)."""
    wrapper = _T5LogitsOnly(model.eval())
    if enc_ids is None:
        enc_ids = torch.randint(0, 100, (1, 8))
    if dec_ids is None:
        dec_ids = torch.randint(0, 100, (1, 6))
    torch.onnx.export(
        wrapper, (enc_ids, dec_ids), output_path,
        input_names=["encoder_input_ids", "decoder_input_ids"],
        output_names=["logits"],
        dynamic_axes={
            "encoder_input_ids": {0: "batch", 1: "enc_seq"},
            "decoder_input_ids": {0: "batch", 1: "dec_seq"},
            "logits": {0: "batch", 1: "dec_seq"},
        },
        opset_version=opset, dynamo=False,
    )
    logger.info(f"exported T5 ONNX model to {output_path}")
    if check:
        try:
            import onnx

            onnx.checker.check_model(onnx.load(output_path))
        except ImportError:
            logger.warning("onnx not installed; skipping checker")
    return output_path


def verify_onnx_model(onnx_path, model, sample_inputs, rtol=1e-2, atol=1e-3):
    """Run the exported graph under onnxruntime and compare against the
    torch model's logits (the reference's check, gpt2_to_onnx.py:26-86).

    ``sample_inputs``: dict of input-name -> int64 tensor.  Raises
    ImportError when onnxruntime is not installed (callers/tests skip)."""
    import numpy as np
    import onnxruntime as ort  # ImportError here is the availability gate

    sess = ort.InferenceSession(onnx_path, providers=["CPUExecutionProvider"])
    feeds = {k: v.numpy() for k, v in sample_inputs.items()}
    (ort_out,) = sess.run(["logits"], feeds)
    with torch.no_grad():
        out = model.eval()(**{k: v for k, v in sample_inputs.items()})
        ref = out["prediction_scores"] if isinstance(out, dict) else out
    np.testing.assert_allclose(ort_out, ref.float().numpy(), rtol=rtol,
                               atol=atol)
    logger.info("onnxruntime outputs match torch within tolerance")
    return True
