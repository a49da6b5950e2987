import pytest
import torch

from replay_amd.nn.utils import gather_last_valid, last_valid_index

pytestmark = pytest.mark.torch


def test_last_valid_index_left_padding():
    mask = torch.tensor([[False, False, True, True], [True, True, True, True]])
    assert last_valid_index(mask).tolist() == [3, 3]


def test_last_valid_index_right_padding():
    mask = torch.tensor([[True, True, False, False], [True, False, False, False]])
    assert last_valid_index(mask).tolist() == [1, 0]


def test_gather_last_valid():
    hidden = torch.arange(24, dtype=torch.float32).reshape(2, 4, 3)
    mask = torch.tensor([[False, True, True, False], [True, True, True, True]])
    out = gather_last_valid(hidden, mask)
    assert out[0].tolist() == hidden[0, 2].tolist()
    assert out[1].tolist() == hidden[1, 3].tolist()
