"""ONNX export + runtime-forward helpers.

Behavioral parity: /root/reference/timm/utils/onnx.py:17 (same export knobs:
dynamic batch/size axes, aten fallback, dynamo path, checker + numeric
round-trip verification).
"""
from typing import List, Optional, Tuple

import torch

__all__ = ['onnx_forward', 'onnx_export']


def onnx_forward(onnx_file, example_input):
    """Run one forward through onnxruntime, returning the first output."""
    import onnxruntime

    session = onnxruntime.InferenceSession(onnx_file, onnxruntime.SessionOptions())
    feed = {session.get_inputs()[0].name: example_input.numpy()}
    return session.run([], feed)[0]


def _resolve_example_input(model, input_size, batch_size, requires_grad):
    if input_size is None:
        assert hasattr(model, 'default_cfg')
        input_size = model.default_cfg.get('input_size')
    return torch.randn((batch_size,) + tuple(input_size), requires_grad=requires_grad)


def onnx_export(
        model: torch.nn.Module,
        output_file: str,
        example_input: Optional[torch.Tensor] = None,
        training: bool = False,
        verbose: bool = False,
        check: bool = True,
        check_forward: bool = False,
        batch_size: int = 64,
        input_size: Tuple[int, int, int] = None,
        opset: Optional[int] = None,
        dynamic_size: bool = False,
        aten_fallback: bool = False,
        keep_initializers: Optional[bool] = None,
        use_dynamo: bool = False,
        input_names: List[str] = None,
        output_names: List[str] = None,
):
    import onnx

    model.train(training)
    if example_input is None:
        example_input = _resolve_example_input(model, input_size, batch_size, training)

    # prime the model once: 'same'-pad layers latch their padding, and the
    # eager output doubles as the dynamo-path verification reference
    eager_out = model(example_input)

    if use_dynamo:
        export_output = torch.onnx.dynamo_export(
            model, example_input,
            export_options=torch.onnx.ExportOptions(dynamic_shapes=dynamic_size),
        )
        export_output.save(output_file)
        torch_out = None
    else:
        input_names = input_names or ['input0']
        output_names = output_names or ['output0']
        dynamic_axes = {'input0': {0: 'batch'}, 'output0': {0: 'batch'}}
        if dynamic_size:
            dynamic_axes['input0'].update({2: 'height', 3: 'width'})
        torch_out = torch.onnx._export(
            model,
            example_input,
            output_file,
            training=(torch.onnx.TrainingMode.TRAINING if training
                      else torch.onnx.TrainingMode.EVAL),
            export_params=True,
            verbose=verbose,
            input_names=input_names,
            output_names=output_names,
            keep_initializers_as_inputs=keep_initializers,
            dynamic_axes=dynamic_axes,
            opset_version=opset,
            operator_export_type=(
                torch.onnx.OperatorExportTypes.ONNX_ATEN_FALLBACK if aten_fallback
                else torch.onnx.OperatorExportTypes.ONNX),
        )

    if check:
        onnx.checker.check_model(onnx.load(output_file), full_check=True)
        if check_forward and not training:
            import numpy as np
            ort_out = onnx_forward(output_file, example_input)
            reference = torch_out if torch_out is not None else eager_out.detach()
            np.testing.assert_almost_equal(reference.numpy(), ort_out, decimal=3)
