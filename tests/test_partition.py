"""Partitioner + meta-device shape inference tests."""

import torch

from mpi4dl_amd.models.resnet import get_resnet_v1, get_resnet_v2
from mpi4dl_amd.parallel.partition import model_generator


def test_even_split_and_balance():
    model = get_resnet_v1((1, 3, 32, 32), n=3)  # 1 stem + 9 blocks + head = 11 cells
    gen = model_generator(model, split_size=4, input_size=(1, 3, 32, 32))
    assert sum(gen.balance) == len(model)
    assert len(gen.bounds) == 4
    # contiguous, ordered, covering
    flat = []
    for s, e in gen.bounds:
        flat.extend(range(s, e))
    assert flat == list(range(len(model)))


def test_explicit_balance():
    model = get_resnet_v1((1, 3, 32, 32), n=3)
    gen = model_generator(
        model, split_size=2, input_size=(1, 3, 32, 32), balance=[3, len(model) - 3]
    )
    assert gen.bounds[0] == (0, 3)


def test_shape_inference_matches_eager():
    model = get_resnet_v2((2, 3, 64, 64), n=2, num_filters=8)
    gen = model_generator(model, split_size=3, input_size=(2, 3, 64, 64))
    shapes = gen.get_output_shapes()
    # check against a real forward
    x = torch.randn(2, 3, 64, 64)
    with torch.no_grad():
        for i, (s, e) in enumerate(gen.bounds):
            for j in range(s, e):
                x = model[j](x)
            assert tuple(x.shape) == tuple(shapes[i]), f"stage {i}"


def test_ready_model_materialises_local_stage():
    model = get_resnet_v1((1, 3, 32, 32), n=1)
    gen = model_generator(model, split_size=2, input_size=(1, 3, 32, 32))
    local = gen.ready_model(0, device=torch.device("cpu"))
    y = local(torch.randn(1, 3, 32, 32))
    assert y.shape[0] == 1
