"""Config-space guards: very-high-resolution shape inference stays
meta-cheap, and invalid configurations fail loudly with actionable
messages (the reference corrupts or hangs silently in these cases)."""

import pytest
import torch


def test_meta_shape_inference_8192():
    """BASELINE config 5 scale: AmoebaNet-D at 8192^2 shape-infers on the
    meta device in seconds (no FLOPs/memory) — the round-2 8192^2 run
    plans against these shapes."""
    from mpi4dl_amd.models.amoebanet import amoebanetd
    from mpi4dl_amd.parallel.partition import model_generator

    with torch.device("meta"):
        m = amoebanetd(1000, 18, 416)
    gen = model_generator(m, 8, input_size=(1, 3, 8192, 8192))
    shapes = gen.get_output_shapes()
    assert len(shapes) == 8
    flat = []
    for s in shapes:
        flat.extend(s if isinstance(s, list) else [s])
    assert all(len(t) == 4 or len(t) == 2 for t in flat)


def test_batch_parts_guard():
    from mpi4dl_amd.comm import Communicator
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator
    from mpi4dl_amd.parallel.pipeline import train_model

    comm = Communicator(split_size=1, backend="gloo")
    m = get_resnet_v1((1, 3, 32, 32), 10, n=1, num_filters=8)
    gen = model_generator(m, 1, input_size=(1, 3, 32, 32))
    gen.get_output_shapes()
    gen.ready_model(0, device=torch.device("cpu"))
    with pytest.raises(AssertionError, match="divide"):
        train_model(gen, 0, 3, 2, comm, device=torch.device("cpu"))


def test_square_slicing_guard():
    from mpi4dl_amd.parallel.spatial import verify_spatial_config

    with pytest.raises(AssertionError, match="square"):
        verify_spatial_config("square", 1024, [2])
    with pytest.raises(AssertionError, match="power of two"):
        verify_spatial_config("vertical", 1000, [2])


def test_balance_sum_guard():
    from mpi4dl_amd.models.resnet import get_resnet_v1
    from mpi4dl_amd.parallel.partition import model_generator

    m = get_resnet_v1((1, 3, 32, 32), 10, n=1, num_filters=8)
    with pytest.raises(AssertionError, match="sums to"):
        model_generator(m, 2, input_size=(1, 3, 32, 32), balance=[1, 1])
