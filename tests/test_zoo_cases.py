"""Every ai-benchmark zoo case constructs and steps on CPU.

The GPU suite exercises only the ResNet-50 flagship cases; this guards the
other 8 model definitions (ResNet-152, VGG-16, DeepLab, LSTM) that
`--cases all` depends on.  Input sizes are shrunk where the case allows it
so CPU time stays small; model architecture is exactly the benchmarked one.
"""
import pytest
import torch

from k8s_device_plugin_amd.models import zoo


@pytest.mark.parametrize("name", list(zoo.CASES))
def test_case_constructs(name):
    case = zoo.CASES[name]
    model = zoo.build(case, torch.device("cpu"))
    n_params = sum(p.numel() for p in model.parameters())
    assert n_params > 1000


@pytest.mark.parametrize("name", ["resnet152_inf", "vgg16_inf",
                                  "deeplab_inf", "lstm_inf",
                                  "resnet152_train", "vgg16_train",
                                  "deeplab_train", "lstm_train"])
@pytest.mark.timeout(300)
def test_case_steps_small_input(name):
    case = zoo.CASES[name]
    if case.input_shape and len(case.input_shape) == 3:
        # shrink image cases: batch 1, 64x64 (stride stacks still valid)
        small = zoo.BenchCase(case.name, case.model_fn, case.phase, 1,
                              (case.input_shape[0], 64, 64),
                              num_classes=getattr(case, "num_classes", 1000),
                              seg=getattr(case, "seg", False))
    else:
        # sequence case (LSTM): shrink batch only
        small = zoo.BenchCase(case.name, case.model_fn, case.phase, 1,
                              case.input_shape,
                              num_classes=getattr(case, "num_classes", 1000),
                              seg=getattr(case, "seg", False))
    model = zoo.build(small, torch.device("cpu"))
    batch = zoo.synthetic_batch(small, torch.device("cpu"))
    opt = (torch.optim.SGD(model.parameters(), lr=0.01)
           if small.phase == "training" else None)
    zoo.step(small, model, batch, opt)  # must not raise
